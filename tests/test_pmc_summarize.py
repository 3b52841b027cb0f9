"""tools/pmc_summarize.py: aggregates rocprofv3 CSVs into per-kernel
summaries (runs on the GPU box; tested here with synthetic CSVs)."""

import json
import sys


def test_summarize_counter_stats_and_trace(tmp_path):
    sys.path.insert(0, "tools")
    from tools.pmc_summarize import main

    d = tmp_path / "out" / "runc"
    d.mkdir(parents=True)
    (d / "1_counter_collection.csv").write_text(
        "Correlation_Id,Dispatch_Id,Kernel_Name,Counter_Name,Counter_Value\n"
        "1,1,fused_sgd_kernel,FETCH_SIZE,100.5\n"
        "2,2,fused_sgd_kernel,FETCH_SIZE,99.5\n"
        "1,1,fused_sgd_kernel,SQ_WAVE_CYCLES,1000\n"
    )
    (d / "1_kernel_stats.csv").write_text(
        '"NAME","CALLS","DURATIONS","AVERAGE","PERCENT","MIN","MAX","STD_DEV","TotalDurationNs"\n'
        '"fused_sgd_kernel",4,x,y,z,1,2,3,4000\n'
    )
    (d / "1_kernel_trace.csv").write_text(
        "Kind,Kernel_Name,Start_Timestamp,End_Timestamp\n"
        "K,fused_sgd_kernel,1000,2000\n"
        "K,fused_sgd_kernel,5000,5500\n"  # second half of the window
        "K,fused_sgd_kernel,9000,9400\n"
    )
    dest = tmp_path / "summary.json"
    rc = main(str(tmp_path / "out"), str(dest))
    assert rc == 0
    s = json.loads(dest.read_text())
    ctr = s["counters"]["fused_sgd_kernel"]
    assert ctr["sums"]["FETCH_SIZE"] == 200.0 and ctr["dispatches"] == 2
    ks = s["kernel_stats"]["fused_sgd_kernel"]
    assert ks["calls"] == 4 and ks["total_ns"] == 4000.0
    ss = s["steady_state"]["fused_sgd_kernel"]
    # mid = 5200: only the last dispatch (dur 400 ns) is steady-state
    assert ss["dispatches"] == 1 and abs(ss["p50_us"] - 0.4) < 1e-9
