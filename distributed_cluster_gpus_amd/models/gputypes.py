"""GPU type specs and startup validation.

Capability parity: reference ``GPUType`` (simcore/models.py:38-46) and
``validate_gpus`` (simcore/validators.py:5-46).
"""
from dataclasses import dataclass
from typing import Iterable, List, Optional


@dataclass(frozen=True)
class GPUSpec:
    name: str
    p_idle: float   # W, powered but idle
    p_peak: float   # W, dynamic part at f=1.0 (added on top of idle)
    p_sleep: float  # W, power-gated (DRS)
    alpha: float = 3.0          # dynamic power ~ f^alpha (baseline model)
    tdp: Optional[float] = None  # declared TDP/TBP (W), for validation only


def validate_gpu_specs(specs: Iterable[GPUSpec], strict: bool = False) -> List[str]:
    """Sanity-check GPU power specs; returns warning strings.

    Checks (parity with reference validate_gpus, simcore/validators.py:5-46):
    negative values, p_sleep > p_idle, alpha outside [1,5], p_idle+p_peak vs TDP
    (both over and << 50% under).  With strict=True, any warning raises.
    """
    msgs: List[str] = []
    seen = set()
    for g in specs:
        if id(g) in seen:
            continue
        seen.add(id(g))
        pre = f"[GPUSpec:{g.name}]"
        if g.p_idle < 0 or g.p_peak < 0 or g.p_sleep < 0:
            msgs.append(f"{pre} negative power value "
                        f"(p_idle={g.p_idle}, p_peak={g.p_peak}, p_sleep={g.p_sleep}).")
        if g.p_sleep > g.p_idle + 1e-6:
            msgs.append(f"{pre} p_sleep ({g.p_sleep} W) > p_idle ({g.p_idle} W); "
                        f"check the configuration/measurements.")
        if not (1.0 <= g.alpha <= 5.0):
            msgs.append(f"{pre} alpha={g.alpha} outside [1,5]; fit it from measured data.")
        if g.tdp is not None:
            total = g.p_idle + g.p_peak
            if total > g.tdp + 1e-6:
                msgs.append(f"{pre} p_idle + p_peak = {total:.1f} W > TDP {g.tdp:.1f} W. "
                            f"Set p_peak ~= (TDP - p_idle) for the baseline model.")
            if total < 0.5 * g.tdp:
                msgs.append(f"{pre} p_idle + p_peak = {total:.1f} W << TDP {g.tdp:.1f} W (<=50%).")
    if strict and msgs:
        raise ValueError("GPU config validation failed:\n" + "\n".join(msgs))
    return msgs
