"""GPU fault detection: amd-smi ECC / device-presence
probing feeding the health monitor. On CPU the detector is exercised
against a stub `amd-smi` whose JSON output the test scripts; the same
parse runs against the real tool on an MI355X (see tests/test_gpu_engine
for the on-hardware sanity check)."""

import json
import os
import stat
import time

import pytest

from agentainer_amd.health.monitor import GpuFaultDetector, HealthMonitor


def _stub_smi(tmp_path, payloads):
    """A fake amd-smi that prints payloads[i] on its i-th invocation."""
    state = tmp_path / "calls"
    state.write_text("0")
    data = tmp_path / "payloads.json"
    data.write_text(json.dumps(payloads))
    script = tmp_path / "amd-smi-stub"
    script.write_text(f"""#!/usr/bin/env python3
import json, sys
state = open({str(state)!r}).read().strip()
i = int(state or 0)
payloads = json.load(open({str(data)!r}))
p = payloads[min(i, len(payloads) - 1)]
open({str(state)!r}, "w").write(str(i + 1))
if p == "__FAIL__":
    sys.exit(3)
print(json.dumps(p))
""")
    script.chmod(script.stat().st_mode | stat.S_IEXEC)
    return [str(script)]


def _ecc(ue, ce):
    # amd-smi metric --ecc --json shape (per-GPU list entries)
    return [{"gpu": 0, "ecc": {"total_correctable_count": ce,
                               "total_uncorrectable_count": ue,
                               "total_deferred_count": 0}}]


def test_detector_ecc_increase_is_fault(tmp_path):
    cmd = _stub_smi(tmp_path, [_ecc(0, 5), _ecc(0, 6), _ecc(2, 6)])
    det = GpuFaultDetector(interval_s=0.0, cmd=cmd)
    assert det.check() is None                 # baseline
    assert det.check() is None                 # correctable creep: no fault
    assert det.status["correctable"] == 6
    assert det.check() == "ecc_uncorrectable"  # UE count increased
    assert det.status["uncorrectable"] == 2


def test_detector_device_disappearance_is_fault(tmp_path):
    cmd = _stub_smi(tmp_path, [_ecc(0, 0), "__FAIL__"])
    det = GpuFaultDetector(interval_s=0.0, cmd=cmd)
    assert det.check() is None
    assert det.check() == "gpu_missing"


def test_detector_missing_binary_is_fault(tmp_path):
    det = GpuFaultDetector(interval_s=0.0,
                           cmd=[str(tmp_path / "no-such-amd-smi")])
    assert det.check() == "gpu_missing"


def test_detector_rate_limit(tmp_path):
    cmd = _stub_smi(tmp_path, [_ecc(0, 0), "__FAIL__"])
    det = GpuFaultDetector(interval_s=3600.0, cmd=cmd)
    now = time.time()
    assert det.check(now) is None
    # within the interval the cached verdict is returned, no subprocess
    assert det.check(now + 1.0) is None
    assert det.check(now + 3601.0) == "gpu_missing"


@pytest.mark.gpu
def test_real_amd_smi_probe_on_hardware():
    """On a real MI355X the stock probe must parse amd-smi's output and
    report a healthy baseline (no fault, >= 1 GPU)."""
    det = GpuFaultDetector(interval_s=0.0)
    fault = det.check()
    assert fault is None, det.status
    assert det.status["ok"] and det.status["n_gpus"] >= 1
    assert det.status["uncorrectable"] >= 0
    # second sample against the baseline: still healthy
    assert det.check() is None


def test_fault_flips_agents_unhealthy(tmp_path):
    """A device fault marks every monitored agent down and persists the
    device status at health:gpu (the monitor.go:267-270 analog)."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.echo import EchoEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    store = Store(str(tmp_path / "state"), sync="never")
    eng = EchoEngine(store)
    man = Manager(store, eng, cfg)
    a = man.deploy(name="gpu-agent", model="echo")
    man.start(a.id)

    cmd = _stub_smi(tmp_path, [_ecc(0, 0), _ecc(1, 0)])
    det = GpuFaultDetector(interval_s=0.0, cmd=cmd)
    mon = HealthMonitor(store, man, retries=3, gpu_fault=det)
    mon.start_monitoring(a.id)

    mon.check_due()                      # baseline: healthy
    st = mon.get_status(a.id)
    assert st["healthy"] is True
    assert store.get("health:gpu")["fault"] is None

    mon._watch[a.id]["next_at"] = 0.0    # force the next probe due
    mon.check_due()                      # UE increased -> device fault
    st = mon.get_status(a.id)
    assert st["healthy"] is False
    assert st["gpu_fault"] == "ecc_uncorrectable"
    assert store.get("health:gpu")["fault"] == "ecc_uncorrectable"
    store.close()
