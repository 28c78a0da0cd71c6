"""{placeholder} templating for jobs and pipeline stages.

Reference parity: llmq/utils/template.py:11-135 (which the reference
partially duplicates inside submit.py:162-236 — unified here).
"""

from __future__ import annotations

import json
import re
from typing import Any, Dict, List, Optional

from llmq_amd.core.models import Job

_PLACEHOLDER_RE = re.compile(r"(?<!\{)\{([A-Za-z_][A-Za-z0-9_]*)\}(?!\})")


def extract_placeholders(template: str) -> List[str]:
    """Names of {placeholders} in a template ({{literal}} braces excluded)."""
    return list(dict.fromkeys(_PLACEHOLDER_RE.findall(template)))


def resolve_template_string(template: str, data: Dict[str, Any]) -> str:
    """Substitute {var} from data; literal {{ }} are preserved as { }.
    Missing variables raise KeyError (validate first with
    validate_required_fields for a friendlier error)."""

    def sub(match: re.Match) -> str:
        key = match.group(1)
        if key not in data:
            raise KeyError(key)
        value = data[key]
        return value if isinstance(value, str) else json.dumps(value, default=str)

    out = _PLACEHOLDER_RE.sub(sub, template)
    return out.replace("{{", "{").replace("}}", "}")


def format_json_template(obj: Any, data: Dict[str, Any]) -> Any:
    """Recursively resolve templates inside dicts/lists/strings (e.g. chat
    messages with {placeholders} in their content)."""
    if isinstance(obj, str):
        return resolve_template_string(obj, data)
    if isinstance(obj, dict):
        return {k: format_json_template(v, data) for k, v in obj.items()}
    if isinstance(obj, list):
        return [format_json_template(v, data) for v in obj]
    return obj


def validate_required_fields(template: str, data: Dict[str, Any]) -> Optional[List[str]]:
    """Return missing placeholder names, or None if all present."""
    missing = [name for name in extract_placeholders(template) if name not in data]
    return missing or None


def create_job_from_data(
    row: Dict[str, Any],
    job_id: str,
    prompt_template: Optional[str] = None,
    column_mapping: Optional[Dict[str, str]] = None,
) -> Job:
    """Build a Job from a data row (JSONL line or dataset item).

    column_mapping renames row columns to template variables
    (``--map template_var=row_column``). If the row itself has prompt or
    messages and no template is given, those are used directly.
    """
    data = dict(row)
    if column_mapping:
        # Three --map value forms, reference parity (submit.py:184-236):
        #   var=[...]        JSON template, recursively interpolated (messages)
        #   var=... {col} ...  template string interpolated against the row
        #   var=column       plain column rename
        import json as _json

        for template_var, mapping_value in column_mapping.items():
            mv = mapping_value.strip()
            if mv.startswith("[") and mv.endswith("]"):
                try:
                    template_obj = _json.loads(mv)
                except _json.JSONDecodeError as exc:
                    raise ValueError(
                        f"--map {template_var}: invalid JSON template: {exc}"
                    ) from exc
                data[template_var] = format_json_template(template_obj, row)
            elif "{" in mv and "}" in mv:
                data[template_var] = resolve_template_string(mv, row)
            elif mapping_value in row:
                data[template_var] = row[mapping_value]
    if prompt_template is not None:
        missing = validate_required_fields(prompt_template, data)
        if missing:
            raise ValueError(
                f"Missing fields for template: {', '.join(missing)} (row keys: {list(row.keys())})"
            )
        extra = {k: v for k, v in data.items() if k not in ("id", "prompt", "messages")}
        return Job(id=job_id, prompt=prompt_template, **extra)
    if "prompt" not in data and "messages" not in data and "text" in data:
        # reference fallback (submit.py:226-229): a bare `text` column
        # becomes the prompt (the common HF-dataset shape)
        data["prompt"] = str(data["text"])
    if "prompt" in data or "messages" in data:
        data.setdefault("id", job_id)
        data["id"] = data.get("id") or job_id
        return Job(**data)
    raise ValueError(
        "Row has neither 'prompt' nor 'messages' and no template was provided"
    )
