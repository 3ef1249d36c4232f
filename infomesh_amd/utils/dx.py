"""Developer-experience generators.

Reference parity: infomesh/dx.py (tool-guide and changelog generators,
custom tokenizer hook surface — the hook itself lives in
utils/plugins.py `tokenizer`/`scorer` slots).
"""
from __future__ import annotations

import subprocess
from pathlib import Path


def generate_tool_guide() -> str:
    """Markdown guide for the MCP tool surface, generated from the live
    schemas (always in sync with mcp/tools.py)."""
    from ..mcp.tools import TOOLS, LEGACY_ALIASES
    lines = ["# infomesh-amd MCP tool guide", ""]
    for tool in TOOLS:
        lines.append(f"## `{tool['name']}`")
        lines.append("")
        lines.append(tool["description"])
        lines.append("")
        props = tool["inputSchema"].get("properties", {})
        required = set(tool["inputSchema"].get("required", []))
        if props:
            lines.append("| argument | type | required |")
            lines.append("|---|---|---|")
            for name, schema in props.items():
                typ = schema.get("type", "any")
                if "enum" in schema:
                    typ += " (" + "|".join(map(str, schema["enum"])) + ")"
                lines.append(
                    f"| `{name}` | {typ} | "
                    f"{'yes' if name in required else 'no'} |")
            lines.append("")
    alias_rows = sorted(LEGACY_ALIASES.items())
    lines.append("## Legacy aliases")
    lines.append("")
    lines.append("| alias | resolves to |")
    lines.append("|---|---|")
    lines.extend(f"| `{a}` | `{t}` |" for a, t in alias_rows)
    lines.append("")
    return "\n".join(lines)


def generate_changelog(repo_dir: Path | None = None,
                       max_entries: int = 50) -> str:
    """Changelog skeleton from git history."""
    cwd = str(repo_dir) if repo_dir else None
    try:
        out = subprocess.run(
            ["git", "log", f"-{max_entries}", "--pretty=%ad|%h|%s",
             "--date=short"],
            capture_output=True, text=True, timeout=10, cwd=cwd)
    except (OSError, subprocess.TimeoutExpired):
        return "# Changelog\n\n(no git history available)\n"
    lines = ["# Changelog", ""]
    current_date = None
    for row in out.stdout.splitlines():
        try:
            date, sha, subject = row.split("|", 2)
        except ValueError:
            continue
        if date != current_date:
            lines.append(f"\n## {date}")
            current_date = date
        lines.append(f"- {subject} (`{sha}`)")
    return "\n".join(lines) + "\n"


def write_docs(out_dir: Path) -> list[Path]:
    out_dir.mkdir(parents=True, exist_ok=True)
    guide = out_dir / "TOOL_GUIDE.md"
    guide.write_text(generate_tool_guide())
    changelog = out_dir / "CHANGELOG.md"
    changelog.write_text(generate_changelog())
    return [guide, changelog]
