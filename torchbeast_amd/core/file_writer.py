"""Experiment metrics/metadata writer (ref: torchbeast/core/file_writer.py).

Creates `{rootdir}/{xpid}/` containing:
- `out.log`     — log messages,
- `logs.csv`    — one row per `log()` tick; the column schema grows
                  dynamically as new metric keys appear,
- `fields.csv`  — history of the column schema,
- `meta.json`   — experiment args + environment metadata,
and maintains a `latest` symlink next to the xpid directory. On re-creation
with an existing xpid the writer appends, continuing the tick counter.
"""

import copy
import csv
import datetime
import json
import logging
import os
import time
from typing import Dict


def _gather_metadata() -> Dict:
    date_start = datetime.datetime.now().isoformat()
    # Git metadata, when running from a checkout.
    git = {}
    try:
        import subprocess

        git["commit"] = (
            subprocess.check_output(
                ["git", "rev-parse", "HEAD"], stderr=subprocess.DEVNULL
            )
            .decode()
            .strip()
        )
    except Exception:
        pass
    slurm = {
        k.split("_", 1)[1].lower(): v
        for k, v in os.environ.items()
        if k.startswith("SLURM_")
    }
    return dict(
        date_start=date_start,
        date_end=None,
        successful=False,
        git=git,
        slurm=slurm or None,
        env=dict(os.environ),
    )


class FileWriter:
    def __init__(self, xpid: str = None, xp_args: dict = None, rootdir: str = "~/palaas"):
        if not xpid:
            xpid = f"{os.getpid()}_{int(time.time())}"
        self.xpid = xpid
        self._tick = 0

        self.metadata = _gather_metadata()
        self.metadata["args"] = copy.deepcopy(xp_args) if xp_args else {}
        self.metadata["xpid"] = self.xpid

        formatter = logging.Formatter("%(message)s")
        self._logger = logging.getLogger(f"palaas/{xpid}")
        self._logger.propagate = False
        self._logger.setLevel(logging.INFO)
        if not self._logger.hasHandlers():
            self._logger.addHandler(logging.StreamHandler())

        rootdir = os.path.expandvars(os.path.expanduser(rootdir))
        self.basepath = os.path.join(rootdir, self.xpid)
        os.makedirs(self.basepath, exist_ok=True)

        # Point {rootdir}/latest at the newest experiment directory.
        symlink = os.path.join(rootdir, "latest")
        try:
            if os.path.islink(symlink):
                os.remove(symlink)
            if not os.path.exists(symlink):
                os.symlink(self.basepath, symlink)
        except OSError:
            pass

        self.paths = dict(
            msg=os.path.join(self.basepath, "out.log"),
            logs=os.path.join(self.basepath, "logs.csv"),
            fields=os.path.join(self.basepath, "fields.csv"),
            meta=os.path.join(self.basepath, "meta.json"),
        )

        self._logger.addHandler(logging.FileHandler(self.paths["msg"]))
        for handler in self._logger.handlers:
            handler.setFormatter(formatter)

        self._save_metadata()

        self.fieldnames = ["_tick", "_time"]
        if os.path.exists(self.paths["logs"]):
            # Resume: recover the schema and continue the tick counter.
            with open(self.paths["logs"], "r") as f:
                reader = csv.reader(f)
                lines = list(reader)
            if len(lines) > 1:
                self.fieldnames = lines[0]
                try:
                    self._tick = int(lines[-1][0]) + 1
                except (ValueError, IndexError):
                    pass

    def log(self, to_log: Dict, tick: int = None, verbose: bool = False) -> None:
        if tick is not None:
            raise NotImplementedError
        to_log["_tick"] = self._tick
        self._tick += 1
        to_log["_time"] = time.time()

        old_len = len(self.fieldnames)
        for k in to_log:
            if k not in self.fieldnames:
                self.fieldnames.append(k)
        if old_len != len(self.fieldnames):
            with open(self.paths["fields"], "a") as f:
                csv.writer(f).writerow(self.fieldnames)
            self._logger.info("Updated log fields: %s", self.fieldnames)

        if to_log["_tick"] == 0:
            with open(self.paths["logs"], "a") as f:
                f.write("# %s\n" % ",".join(self.fieldnames))

        if verbose:
            self._logger.info(
                "LOG | %s",
                ", ".join(f"{k}: {to_log[k]}" for k in sorted(to_log)),
            )

        with open(self.paths["logs"], "a") as f:
            writer = csv.DictWriter(f, fieldnames=self.fieldnames)
            writer.writerow(to_log)

    def close(self, successful: bool = True) -> None:
        self.metadata["date_end"] = datetime.datetime.now().isoformat()
        self.metadata["successful"] = successful
        self._save_metadata()

    def _save_metadata(self) -> None:
        with open(self.paths["meta"], "w") as f:
            json.dump(self.metadata, f, indent=4, sort_keys=True, default=str)
