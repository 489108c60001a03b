"""Straggler detection + GPU telemetry on MI355X.

Capability analog of reference megatron/core/utils.py:1445 StragglerDetector
(CUDA-event step timing + NVML power/temp, min/max-rank reports) — telemetry
comes from the amdgpu sysfs hwmon interface (what rocm-smi reads) instead of
pynvml, and timing uses hipEvents through torch.cuda.Event.
"""

from __future__ import annotations

import glob
import os
import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.distributed as dist


def _read_int(path: str) -> Optional[int]:
    try:
        with open(path) as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return None


class AmdGpuTelemetry:
    """Power (W), temperature (C), and sclk for the local GPU via sysfs."""

    def __init__(self, device_index: int = 0):
        hwmons = sorted(glob.glob("/sys/class/drm/card*/device/hwmon/hwmon*"))
        self._hw = hwmons[device_index] if device_index < len(hwmons) else None

    def power_w(self) -> Optional[float]:
        if not self._hw:
            return None
        for name in ("power1_average", "power1_input"):
            v = _read_int(os.path.join(self._hw, name))
            if v is not None:
                return v / 1e6
        return None

    def temp_c(self) -> Optional[float]:
        if not self._hw:
            return None
        v = _read_int(os.path.join(self._hw, "temp1_input"))
        return v / 1000.0 if v is not None else None

    def energy_j(self) -> Optional[float]:
        if not self._hw:
            return None
        v = _read_int(os.path.join(self._hw, "energy1_input"))
        return v / 1e6 if v is not None else None


@dataclass
class StragglerReport:
    min_rank: int
    max_rank: int
    min_time_ms: float
    max_time_ms: float
    mean_time_ms: float
    power_w: Optional[float] = None
    temp_c: Optional[float] = None


class StragglerDetector:
    """Times a per-rank section each step; reports min/max ranks on demand."""

    def __init__(self, enabled: bool = False, device_index: int = 0,
                 control_port: Optional[int] = None):
        self.enabled = enabled
        self._use_events = torch.cuda.is_available()
        self._start_evt = None
        self._t0 = 0.0
        self._elapsed: List[float] = []
        self.telemetry = AmdGpuTelemetry(device_index)
        self._server = None
        if control_port is not None:
            self._start_control_server(control_port)

    def _start_control_server(self, port: int):
        """Runtime on/off toggle via `curl host:<port>` (reference
        StragglerDetector's port-toggled control, core/utils.py:1445)."""
        import socket
        import threading

        srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind(("127.0.0.1", port))
        srv.listen(4)
        self._server = srv
        self.control_port = srv.getsockname()[1]

        def serve():
            while True:
                try:
                    conn, _ = srv.accept()
                except OSError:
                    return  # closed
                with conn:
                    try:
                        conn.recv(1024)
                        self.enabled = not self.enabled
                        state = b"on" if self.enabled else b"off"
                        conn.sendall(b"HTTP/1.0 200 OK\r\n\r\nstraggler detection " + state + b"\r\n")
                    except OSError:
                        pass

        threading.Thread(target=serve, daemon=True).start()

    def close(self):
        if self._server is not None:
            try:
                self._server.close()
            except OSError:
                pass
            self._server = None

    def start(self):
        if not self.enabled:
            return
        if self._use_events:
            self._start_evt = torch.cuda.Event(enable_timing=True)
            self._end_evt = torch.cuda.Event(enable_timing=True)
            self._start_evt.record()
        else:
            self._t0 = time.perf_counter()

    def stop(self):
        if not self.enabled:
            return
        if self._use_events and self._start_evt is not None:
            self._end_evt.record()
            self._end_evt.synchronize()
            self._elapsed.append(self._start_evt.elapsed_time(self._end_evt))
        else:
            self._elapsed.append((time.perf_counter() - self._t0) * 1e3)

    def report(self, group=None) -> Optional[StragglerReport]:
        """All-gather per-rank mean section time; returns the report on rank 0
        (None elsewhere or when disabled/no data). Resets the window."""
        if not self.enabled or not self._elapsed:
            return None
        mine = sum(self._elapsed) / len(self._elapsed)
        self._elapsed.clear()
        if dist.is_initialized():
            world = dist.get_world_size(group)
            t = torch.tensor([mine], dtype=torch.float64)
            all_t = [torch.zeros_like(t) for _ in range(world)]
            dist.all_gather(all_t, t, group=group)
            times = [float(x) for x in all_t]
            if dist.get_rank(group) != 0:
                return None
        else:
            times = [mine]
        mn, mx = min(times), max(times)
        return StragglerReport(
            min_rank=times.index(mn), max_rank=times.index(mx),
            min_time_ms=mn, max_time_ms=mx, mean_time_ms=sum(times) / len(times),
            power_w=self.telemetry.power_w(), temp_c=self.telemetry.temp_c())


class EnergyMonitor:
    """Per-interval GPU energy (J) via the hwmon energy counter, with a
    power-integration fallback (reference core/energy_monitor.py analog)."""

    def __init__(self, device_index: int = 0):
        self.telemetry = AmdGpuTelemetry(device_index)
        self._last_e = None
        self._last_t = None

    def lap(self) -> Optional[float]:
        e = self.telemetry.energy_j()
        now = time.perf_counter()
        out = None
        if e is not None and self._last_e is not None:
            out = e - self._last_e
        elif self._last_t is not None:
            p = self.telemetry.power_w()
            if p is not None:
                out = p * (now - self._last_t)
        self._last_e, self._last_t = e, now
        return out
