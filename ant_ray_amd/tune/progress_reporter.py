"""Progress reporters.

Role parity: reference python/ray/tune/progress_reporter.py
(ProgressReporter base, CLIReporter, JupyterNotebookReporter). The Tuner
calls report() whenever trial states change; CLIReporter prints a compact
table of trial status + last metrics.
"""
import sys
import time
from typing import Dict, List, Optional


class ProgressReporter:
    def __init__(self, metric_columns: Optional[List[str]] = None,
                 max_report_frequency: float = 5.0, **_):
        self._metric_columns = metric_columns
        self._freq = max_report_frequency
        self._last = 0.0

    def should_report(self, trials, done: bool = False) -> bool:
        if done or time.time() - self._last >= self._freq:
            self._last = time.time()
            return True
        return False

    def report(self, trials: List[Dict], done: bool, *args):
        raise NotImplementedError


class CLIReporter(ProgressReporter):
    def report(self, trials, done, *args):
        counts: Dict[str, int] = {}
        for t in trials:
            counts[t.get("status", "?")] = counts.get(t.get("status", "?"), 0) + 1
        line = (f"== Tune status: {len(trials)} trials "
                + " ".join(f"{k}={v}" for k, v in sorted(counts.items())))
        rows = []
        for t in trials:
            m = t.get("metrics") or {}
            cols = self._metric_columns or [
                k for k in m if isinstance(m[k], (int, float))][:4]
            rows.append("  " + t.get("trial_id", "?") + " [" +
                        t.get("status", "?") + "] " +
                        " ".join(f"{c}={m.get(c)}" for c in cols))
        print("\n".join([line] + rows), file=sys.stderr)


class JupyterNotebookReporter(CLIReporter):
    """Same output; notebooks render stderr inline."""
