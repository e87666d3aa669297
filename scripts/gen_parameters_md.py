"""Regenerate docs/PARAMETERS.md from the live config module (run after
adding settings; tests/test_conventions.py keeps them in sync)."""
import importlib
import os
import re
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import audiomuse_amd.config as C  # noqa: E402

importlib.reload(C)
src = open('audiomuse_amd/config.py').read()
lines = src.split('\n')
sections = [("General", [])]
for i, ln in enumerate(lines):
    if re.match(r'^# ---+$', ln) and i + 1 < len(lines) and \
            lines[i + 1].startswith('# '):
        sections.append((lines[i + 1][2:].strip(), []))
    m = re.match(r'^([A-Z][A-Z0-9_]*) *=', ln)
    if m:
        sections[-1][1].append(m.group(1))
SECRET = {"JWT_SECRET", "API_TOKEN", "POSTGRES_PASSWORD",
          "AUDIOMUSE_PASSWORD", "NAVIDROME_PASSWORD", "NAVIDROME_API_KEY",
          "JELLYFIN_TOKEN", "EMBY_TOKEN", "PLEX_TOKEN", "AI_API_KEY",
          "OPENAI_API_KEY", "GEMINI_API_KEY", "MISTRAL_API_KEY"}
out = ["# Configuration parameters", "",
       "Every setting below is an env-read constant in "
       "`audiomuse_amd/config.py`",
       "(reference analog: `/root/reference/config.py` + "
       "`docs/PARAMETERS.md`).",
       "All values can also be persisted through the Setup Wizard / "
       "`POST /api/config` into the `app_config` table; DB overrides are",
       "re-applied at runtime via `refresh_config()` and hydrated by "
       "workers",
       "per job. `DATABASE_URL`/`POSTGRES_*` and `TZ` are env-only, as in "
       "the",
       "reference (docs/ALGORITHM.md:129-135).", ""]
n = 0
for title, names in sections:
    if not names:
        continue
    out += [f"## {title}", "", "| Parameter | Default |", "|---|---|"]
    for name in names:
        v = getattr(C, name, None)
        if isinstance(v, list):
            d = f"({len(v)} entries, code-owned)"
        elif name in SECRET:
            d = "*(secret; empty)*" if not v or v == "no-key-needed" \
                else "*(secret)*"
        else:
            d = f"`{v!r}`"
        out.append(f"| `{name}` | {d} |")
        n += 1
    out.append("")
out.append(f"**Total: {n} parameters.**")
open('docs/PARAMETERS.md', 'w').write('\n'.join(out) + '\n')
print("wrote", n, "parameters")
