"""_ReportAccessor: append/read per-iteration MaterializedReports as JSON.

Mirrors reference adanet/core/report_accessor.py:87-159 (the reference
moved from proto to a JSON-lines file in 0.9; we keep the same
<report_dir>/iteration_reports.json layout with one JSON list per line
keyed by iteration).
"""

from __future__ import annotations

import json
import os
from typing import Dict, Iterable, List

from adanet_amd.subnetwork.report import MaterializedReport


class _ReportAccessor(object):

    def __init__(self, report_dir: str):
        self._report_dir = report_dir
        os.makedirs(report_dir, exist_ok=True)
        self._path = os.path.join(report_dir, "iteration_reports.json")

    @property
    def report_file_path(self) -> str:
        return self._path

    def write_iteration_report(self, iteration_number: int,
                               materialized_reports: Iterable[MaterializedReport]):
        """Appends reports for one iteration (idempotent per iteration:
        re-writing an iteration replaces its entry, matching the reference's
        overwrite-on-retrain behavior)."""
        existing = self._read_raw()
        existing[str(iteration_number)] = [
            r.to_json() for r in materialized_reports
        ]
        tmp = self._path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(existing, f, sort_keys=True)
        os.replace(tmp, self._path)

    def read_iteration_reports(self) -> List[List[MaterializedReport]]:
        """Returns reports grouped by iteration, ordered by iteration."""
        raw = self._read_raw()
        out = []
        for key in sorted(raw, key=int):
            out.append([MaterializedReport.from_json(d) for d in raw[key]])
        return out

    def _read_raw(self) -> Dict[str, list]:
        if not os.path.exists(self._path):
            return {}
        with open(self._path) as f:
            content = f.read().strip()
        if not content:
            return {}
        return json.loads(content)
