"""CPU parity of the DEVICE ALGORITHM: the host simulator (scan_host_sim.cc
compiles scan_device.h — the exact per-interval code the gfx950 kernel runs)
against the CPU oracle on every shared parity scenario. Catches kernel-logic
bugs without a GPU; the gpu-marked suite then validates the same scenarios on
real hardware."""
import pytest

import ybgpu as y
from parity_cases import build_cases, make_spec, run_oracle, check_match


@pytest.fixture(scope="module")
def cases():
    return build_cases()


def _ids():
    # stable ids without building cases at collection time
    return None


def test_sim_vs_oracle_all_cases(cases):
    for case in cases:
        for run in case["runs"]:
            read_micros, preds, aggs = run[0], run[1], run[2]
            lower = run[3] if len(run) > 3 else None
            upper = run[4] if len(run) > 4 else None
            spec = make_spec(case, read_micros, preds, aggs, lower, upper)
            sres = y.sim_scan(spec, case["data"], case["offsets"],
                              case["n_blocks"])
            ores = run_oracle(case, read_micros, preds, aggs, lower, upper)
            try:
                check_match(sres, ores, aggs, check_entries=lower is None and upper is None)
            except AssertionError as e:
                raise AssertionError(
                    f"case {case['name']} read={read_micros}: {e}") from e
