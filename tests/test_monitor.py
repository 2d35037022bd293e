"""GPU monitor poller (statistics.sh parity) with a stubbed rocm-smi."""

import os
import stat
import time


def test_monitor_csv(tmp_path, monkeypatch):
    stub = tmp_path / "rocm-smi"
    stub.write_text(
        "#!/bin/sh\n"
        "echo 'device,GPU use (%),Memory use (%),VRAM Total Memory (B),"
        "VRAM Total Used Memory (B)'\n"
        "echo 'card0,42,17,309237645312,1073741824'\n")
    stub.chmod(stub.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("PATH", f"{tmp_path}{os.pathsep}{os.environ['PATH']}")

    from amdtrain.utils.monitor import GpuMonitor, _query_rocm_smi
    rows = _query_rocm_smi()
    assert rows and rows[0][1] == "0"
    assert rows[0][2] == "309237645312"
    assert rows[0][5] == "42"

    csv_path = tmp_path / "run_log.csv"
    with GpuMonitor(str(csv_path), interval_s=0.05):
        time.sleep(0.3)
    lines = csv_path.read_text().strip().splitlines()
    assert lines[0].startswith("timestamp,index,memory.total")
    assert len(lines) >= 2  # at least one sample row
