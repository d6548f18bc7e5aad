"""Experiment metrics/metadata writer (ref: torchbeast/core/file_writer.py).

One directory per experiment id under `rootdir`:

    {rootdir}/{xpid}/
        out.log      stream of human-readable messages
        logs.csv     one row per log() call, tick-indexed; columns grow
                     as new metric names show up
        fields.csv   one row per schema revision (audit trail of columns)
        meta.json    experiment args + host/git/scheduler metadata
    {rootdir}/latest -> the most recently created xpid directory

Re-opening an existing xpid appends instead of truncating and resumes the
tick counter from the last row, so a preempted-and-restarted run produces
one continuous CSV.
"""

import copy
import csv
import datetime
import json
import logging
import os
import subprocess
import time
from typing import Dict, List, Optional


def _git_commit() -> Optional[str]:
    try:
        out = subprocess.run(
            ["git", "rev-parse", "HEAD"],
            capture_output=True,
            timeout=5,
            check=True,
        )
        return out.stdout.decode().strip()
    except Exception:
        return None


def _scheduler_metadata() -> Optional[Dict[str, str]]:
    """SLURM job context, if any (lowercased keys without the prefix)."""
    found = {}
    for key, value in os.environ.items():
        if not key.startswith("SLURM_"):
            continue
        found[key[len("SLURM_"):].lower()] = value
    return found if found else None


class FileWriter:
    def __init__(self, xpid: str = None, xp_args: dict = None, rootdir: str = "~/palaas"):
        self.xpid = xpid or f"{os.getpid()}_{int(time.time())}"
        self._tick = 0

        self.metadata = {
            "xpid": self.xpid,
            "args": copy.deepcopy(xp_args) if xp_args else {},
            "date_start": datetime.datetime.now().isoformat(),
            "date_end": None,
            "successful": False,
            "git": {"commit": _git_commit()},
            "slurm": _scheduler_metadata(),
            "env": dict(os.environ),
        }

        rootdir = os.path.expandvars(os.path.expanduser(rootdir))
        self.basepath = os.path.join(rootdir, self.xpid)
        os.makedirs(self.basepath, exist_ok=True)
        self._repoint_latest_symlink(rootdir)

        self.paths = {
            name: os.path.join(self.basepath, filename)
            for name, filename in (
                ("msg", "out.log"),
                ("logs", "logs.csv"),
                ("fields", "fields.csv"),
                ("meta", "meta.json"),
            )
        }

        self._logger = logging.getLogger(f"palaas/{self.xpid}")
        self._logger.propagate = False
        self._logger.setLevel(logging.INFO)
        if not self._logger.hasHandlers():
            self._logger.addHandler(logging.StreamHandler())
        self._logger.addHandler(logging.FileHandler(self.paths["msg"]))
        plain = logging.Formatter("%(message)s")
        for handler in self._logger.handlers:
            handler.setFormatter(plain)

        self._save_metadata()
        self.fieldnames = self._recover_schema()

    def _repoint_latest_symlink(self, rootdir: str) -> None:
        link = os.path.join(rootdir, "latest")
        try:
            if os.path.islink(link):
                os.remove(link)
            if not os.path.exists(link):
                os.symlink(self.basepath, link)
        except OSError:
            pass  # e.g. filesystems without symlink support

    def _recover_schema(self) -> List[str]:
        """On resume, re-read logs.csv for the column list and last tick."""
        default = ["_tick", "_time"]
        if not os.path.exists(self.paths["logs"]):
            return default
        with open(self.paths["logs"], "r") as f:
            lines = list(csv.reader(f))
        if len(lines) < 2:
            return default
        try:
            self._tick = int(lines[-1][0]) + 1
        except (ValueError, IndexError):
            pass
        return lines[0]

    def log(self, to_log: Dict, tick: int = None, verbose: bool = False) -> None:
        if tick is not None:
            raise NotImplementedError("explicit ticks are not supported")
        to_log["_tick"] = self._tick
        to_log["_time"] = time.time()
        self._tick += 1

        new_columns = [k for k in to_log if k not in self.fieldnames]
        if new_columns:
            self.fieldnames.extend(new_columns)
            with open(self.paths["fields"], "a") as f:
                csv.writer(f).writerow(self.fieldnames)
            self._logger.info("Updated log fields: %s", self.fieldnames)

        if to_log["_tick"] == 0:
            # Header row, written once. Commented so pandas/np loaders that
            # skip '#' lines and the resume parser both cope.
            with open(self.paths["logs"], "a") as f:
                f.write("# %s\n" % ",".join(self.fieldnames))

        if verbose:
            rendered = ", ".join(f"{k}: {to_log[k]}" for k in sorted(to_log))
            self._logger.info("LOG | %s", rendered)

        with open(self.paths["logs"], "a") as f:
            csv.DictWriter(f, fieldnames=self.fieldnames).writerow(to_log)

    def close(self, successful: bool = True) -> None:
        self.metadata["date_end"] = datetime.datetime.now().isoformat()
        self.metadata["successful"] = successful
        self._save_metadata()

    def _save_metadata(self) -> None:
        with open(self.paths["meta"], "w") as f:
            json.dump(self.metadata, f, indent=4, sort_keys=True, default=str)
