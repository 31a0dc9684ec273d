"""ReportMaterializer (reference adanet/core/report_materializer.py:30-160).

Evaluates subnetwork Report tensors over a few batches and produces
MaterializedReports for the next iteration's Generator. In define-by-run
the reports' tensor values are already concrete at build time, so
materialization reduces to snapshotting; the input_fn/steps arguments are
kept for API parity and for reports whose metrics are callables over data.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

from adanet_amd.subnetwork.report import MaterializedReport, Report


class ReportMaterializer(object):
    """Materializes each candidate's `subnetwork.Report` over `steps`
    batches of `input_fn` into python `MaterializedReport`s for
    report-driven Generators (reference report_materializer.py:30-74)."""

    def __init__(self, input_fn, steps: Optional[int] = None):
        self._input_fn = input_fn
        self._steps = steps

    @property
    def input_fn(self):
        return self._input_fn

    @property
    def steps(self):
        return self._steps

    def materialize_subnetwork_reports(
            self, iteration_number: int, reports: Sequence[Report],
            names: Sequence[str],
            included_in_final_ensemble: Sequence[bool]
    ) -> List[MaterializedReport]:
        out = []
        for report, name, included in zip(reports, names,
                                          included_in_final_ensemble):
            out.append(
                report.materialize(iteration_number, name,
                                   included_in_final_ensemble=included))
        return out
