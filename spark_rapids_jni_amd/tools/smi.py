"""GPU telemetry (reference parity: NVMLJni.cpp + nvml/ Java classes —
device info/clocks/power/temp/memory/utilization structs and the NVMLMonitor
poller). MI355X equivalent: amdsmi python bindings when importable, else the
`amd-smi`/`rocm-smi` CLI."""
import json
import subprocess
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional


@dataclass
class DeviceTelemetry:
    index: int
    name: str = ""
    temperature_c: Optional[float] = None
    power_w: Optional[float] = None
    sclk_mhz: Optional[float] = None
    mclk_mhz: Optional[float] = None
    gfx_busy_percent: Optional[float] = None
    vram_used_mb: Optional[float] = None
    vram_total_mb: Optional[float] = None
    raw: dict = field(default_factory=dict)


def _try_amdsmi() -> Optional[List[DeviceTelemetry]]:
    try:
        import amdsmi
    except ImportError:
        return None
    try:
        amdsmi.amdsmi_init()
        out = []
        for i, h in enumerate(amdsmi.amdsmi_get_processor_handles()):
            t = DeviceTelemetry(index=i)
            try:
                info = amdsmi.amdsmi_get_gpu_asic_info(h)
                t.name = info.get("market_name", "")
            except Exception:
                pass
            try:
                t.power_w = amdsmi.amdsmi_get_power_info(h).get(
                    "average_socket_power")
            except Exception:
                pass
            out.append(t)
        amdsmi.amdsmi_shut_down()
        return out
    except Exception:
        return None


def _cli_json(cmd):
    try:
        r = subprocess.run(cmd, capture_output=True, text=True, timeout=30)
        if r.returncode != 0:
            return None
        return json.loads(r.stdout)
    except Exception:
        return None


def get_telemetry() -> List[DeviceTelemetry]:
    """Snapshot for every visible GPU."""
    res = _try_amdsmi()
    if res:
        return res
    doc = _cli_json(["rocm-smi", "--showtemp", "--showpower", "--showclocks",
                     "--showuse", "--showmeminfo", "vram", "--json"])
    out = []
    if doc:
        for key, card in sorted(doc.items()):
            if not key.startswith("card"):
                continue
            idx = int(key.replace("card", ""))
            t = DeviceTelemetry(index=idx, raw=card)

            def num(*names):
                for nm in names:
                    for k, v in card.items():
                        if nm.lower() in k.lower():
                            try:
                                return float(str(v).split("(")[0])
                            except ValueError:
                                pass
                return None

            t.temperature_c = num("Temperature (Sensor junction)",
                                  "Temperature (Sensor edge)")
            t.power_w = num("Average Graphics Package Power",
                            "Current Socket Graphics Package Power")
            t.gfx_busy_percent = num("GPU use")
            t.vram_used_mb = num("VRAM Total Used Memory")
            t.vram_total_mb = num("VRAM Total Memory")
            out.append(t)
    return out


class SMIMonitor:
    """NVMLMonitor analog: background poller invoking a callback."""

    def __init__(self, period_s: float = 1.0,
                 callback: Optional[Callable[[List[DeviceTelemetry]], None]] = None):
        self.period_s = period_s
        self.callback = callback
        self.samples: List[List[DeviceTelemetry]] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self):
        self._stop.clear()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def _run(self):
        while not self._stop.is_set():
            snap = get_telemetry()
            self.samples.append(snap)
            if self.callback:
                self.callback(snap)
            self._stop.wait(self.period_s)

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=10)
