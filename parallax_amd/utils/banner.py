"""Boot banner + environment report (reference UX analogue:
parallax_utils/ascii_anime.py + version_check.py — the offline environment has
no version-check endpoint, so this reports the local stack instead)."""

from __future__ import annotations

__version__ = "0.2.0"

_BANNER = r"""
  ___  __ _ _ __ __ _| | | __ ___  __      __ _ _ __ ___   __| |
 / _ \/ _` | '__/ _` | | |/ _` \ \/ /____ / _` | '_ ` _ \ / _` |
| (_) | (_| | | | (_| | | | (_| |>  <____| (_| | | | | | | (_| |
 \___/\__,_|_|  \__,_|_|_|\__,_/_/\_\     \__,_|_| |_| |_|\__,_|
  p                                  MI355X-native inference engine
"""


def print_banner(role: str = "serve") -> None:
    import torch

    lines = [_BANNER, f"  v{__version__} · role: {role}"]
    try:
        if torch.cuda.is_available():
            props = torch.cuda.get_device_properties(0)
            lines.append(
                f"  {torch.cuda.device_count()}x {props.name} · "
                f"{props.total_memory / (1 << 30):.0f} GB · "
                f"{props.multi_processor_count} CUs · ROCm/HIP {torch.version.hip}"
            )
        else:
            lines.append("  CPU mode (no ROCm device visible)")
        lines.append(f"  torch {torch.__version__}")
    except Exception:
        pass
    print("\n".join(lines), flush=True)
