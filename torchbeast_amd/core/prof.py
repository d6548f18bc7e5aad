"""Per-phase wall-time statistics for hot loops (capability parity with the
reference's Timings helper, ref: torchbeast/core/prof.py).

Usage: call `time(name)` after each phase of a loop body; the elapsed time
since the previous call is folded into an online mean/variance (Welford's
algorithm, one (count, mean, M2) triple per phase name). `summary()` renders
the phases sorted by cost with their share of the total.
"""

import timeit


class Timings:
    def __init__(self):
        # name -> [count, mean, M2]  (Welford accumulator)
        self._acc = {}
        self.reset()

    def reset(self):
        self.last_time = timeit.default_timer()

    def time(self, name):
        now = timeit.default_timer()
        sample = now - self.last_time
        self.last_time = now

        entry = self._acc.setdefault(name, [0, 0.0, 0.0])
        entry[0] += 1
        delta = sample - entry[1]
        entry[1] += delta / entry[0]
        entry[2] += delta * (sample - entry[1])

    def means(self):
        return {name: e[1] for name, e in self._acc.items()}

    def vars(self):
        return {
            name: (e[2] / e[0] if e[0] else 0.0) for name, e in self._acc.items()
        }

    def stds(self):
        return {name: v**0.5 for name, v in self.vars().items()}

    def summary(self, prefix=""):
        means = self.means()
        stds = self.stds()
        total = sum(means.values())

        lines = [prefix]
        for name in sorted(means, key=means.get, reverse=True):
            share = 100 * means[name] / total if total else 0.0
            lines.append(
                f"    {name}: {1000 * means[name]:.6f}ms "
                f"+- {1000 * stds[name]:.6f}ms ({share:.2f}%) "
            )
        lines.append(f"Total: {1000 * total:.6f}ms")
        return "\n".join(lines)
