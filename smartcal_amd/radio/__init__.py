"""smartcal_amd.radio — the radio-astronomy math layer (MI355X-native).

Re-designs the reference's ``calibration/calibration_tools.py`` (L2 of
SURVEY.md §1) as batched, device-resident tensor math: what the reference
computes with quadruple Python loops per (direction, timeslot, baseline)
(`calibration_tools.py:589-1178`) is expressed here as fused einsum /
index_add batches that run as a handful of GPU kernels, plus host-side
setup math (consensus polynomials) that is not hot.

Data lives in memory (HBM3E) instead of MeasurementSets and text files;
parsers/writers for the reference's text formats are provided for
compatibility (`smartcal_amd.radio.sky`, `smartcal_amd.radio.solutions`).
"""

from . import (coords, sky, coherency, consensus, hessian, solutions,  # noqa: F401
               array, sim, solver, imaging, influence, shapelet,  # noqa: F401
               dataset, io, small_complex)  # noqa: F401
