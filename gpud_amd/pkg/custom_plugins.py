"""Custom plugins: user-defined bash health checks as components.

Reference: pkg/custom-plugins/types.go:36-170 — Spec{plugin_type:
init|component|component_list, run_mode auto|manual, tags,
health_state_plugin.steps[].run_bash_script, parser.json_paths[] with
expect match-rules and suggested-action mappings, timeout, interval};
specs load from YAML, become registry components named
``custom-plugin-<name>`` (server wiring reference: pkg/server/
server.go:351-389 — init-type plugins run once and gate daemon start).

The output parser supports dot-path queries into the last line of JSON
output (the reference uses jq-style paths; dot paths cover the documented
examples) with ``expect.regex`` rules and per-rule suggested actions.
"""

from __future__ import annotations

import base64
import json
import os
import re
from dataclasses import dataclass
from dataclasses import field as dc_field
from typing import Any, Dict, List, Optional

import yaml

from ..apiv1.types import (
    ComponentType,
    HealthStateType,
    RunModeType,
    SuggestedActions,
)
from ..components.base import CheckResult, Component, TickerComponent
from .process_runner import run_bash

PLUGIN_TYPE_INIT = "init"
PLUGIN_TYPE_COMPONENT = "component"
PLUGIN_TYPE_COMPONENT_LIST = "component_list"

NAME_PREFIX = "custom-plugin-"


@dataclass
class RunBashScript:
    script: str = ""
    content_type: str = "plaintext"  # or base64

    def decoded(self) -> str:
        if self.content_type == "base64":
            return base64.b64decode(self.script).decode()
        return self.script

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "RunBashScript":
        return RunBashScript(
            script=d.get("script", ""),
            content_type=d.get("content_type", "plaintext"),
        )


@dataclass
class JSONPathRule:
    query: str = ""  # dot path, e.g. "result" or "data.status"
    field: str = ""
    expect_regex: str = ""
    suggested_actions: Dict[str, str] = dc_field(default_factory=dict)  # action -> regex

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "JSONPathRule":
        expect = d.get("expect") or {}
        return JSONPathRule(
            query=d.get("query", ""),
            field=d.get("field", ""),
            expect_regex=expect.get("regex", ""),
            suggested_actions=d.get("suggested_actions") or {},
        )


@dataclass
class Spec:
    plugin_name: str = ""
    plugin_type: str = PLUGIN_TYPE_COMPONENT
    run_mode: str = RunModeType.AUTO
    tags: List[str] = dc_field(default_factory=list)
    steps: List[RunBashScript] = dc_field(default_factory=list)
    json_paths: List[JSONPathRule] = dc_field(default_factory=list)
    # optional plugin-output log with ${PLUGIN}/${TRIGGER} substitution
    # (reference: docs/PLUGIN.md "Log Path Variable Substitution")
    log_path: str = ""
    timeout_seconds: float = 60.0
    interval_seconds: float = 600.0
    component_list: List[str] = dc_field(default_factory=list)
    component_list_file: str = ""

    @property
    def component_name(self) -> str:
        return NAME_PREFIX + self.plugin_name

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Spec":
        hsp = d.get("health_state_plugin") or {}
        steps = [
            RunBashScript.from_dict(s.get("run_bash_script") or {})
            for s in (hsp.get("steps") or [])
        ]
        parser = hsp.get("parser") or {}
        rules = [JSONPathRule.from_dict(r) for r in (parser.get("json_paths") or [])]
        timeout = d.get("timeout", "60s")
        interval = d.get("interval", "10m")
        return Spec(
            plugin_name=d.get("plugin_name", ""),
            plugin_type=d.get("plugin_type", PLUGIN_TYPE_COMPONENT),
            run_mode=d.get("run_mode", RunModeType.AUTO) or RunModeType.AUTO,
            tags=list(d.get("tags") or []),
            steps=steps,
            json_paths=rules,
            log_path=str(parser.get("log_path") or ""),
            timeout_seconds=parse_duration(timeout),
            interval_seconds=parse_duration(interval),
            component_list=list(d.get("component_list") or []),
            component_list_file=str(d.get("component_list_file") or ""),
        )

    def to_dict(self) -> Dict[str, Any]:
        return {
            "plugin_name": self.plugin_name,
            "plugin_type": self.plugin_type,
            "run_mode": self.run_mode,
            "tags": list(self.tags),
            "health_state_plugin": {
                "steps": [
                    {
                        "run_bash_script": {
                            "script": s.script,
                            "content_type": s.content_type,
                        }
                    }
                    for s in self.steps
                ],
                "parser": {
                    "json_paths": [
                        {
                            "query": r.query,
                            "field": r.field,
                            "expect": {"regex": r.expect_regex}
                            if r.expect_regex
                            else {},
                            "suggested_actions": r.suggested_actions,
                        }
                        for r in self.json_paths
                    ]
                },
            },
            "timeout": f"{self.timeout_seconds:g}s",
            "interval": f"{self.interval_seconds:g}s",
        }

    def validate(self) -> Optional[str]:
        if not self.plugin_name:
            return "plugin_name is required"
        if self.plugin_type not in (
            PLUGIN_TYPE_INIT,
            PLUGIN_TYPE_COMPONENT,
            PLUGIN_TYPE_COMPONENT_LIST,
        ):
            return f"invalid plugin_type {self.plugin_type!r}"
        if not self.steps:
            return "at least one step is required"
        if self.plugin_type == PLUGIN_TYPE_COMPONENT_LIST and not (
            self.component_list or self.component_list_file
        ):
            return "component_list or component_list_file is required"
        return None


def parse_duration(v: Any) -> float:
    """'90s' / '10m' / '1h' / numeric seconds → seconds (defaults to 60 on
    unparsable input; scientific-notation numerics from YAML round-trips
    are accepted)."""
    if isinstance(v, (int, float)):
        return float(v)
    s = str(v).strip()
    m = re.fullmatch(r"([0-9.eE+-]+)\s*(ms|s|m|h)?", s)
    if not m:
        return 60.0
    try:
        val = float(m.group(1))
    except ValueError:
        return 60.0
    if val < 0:
        return 60.0
    return val * {"ms": 0.001, "s": 1, "m": 60, "h": 3600, None: 1}[m.group(2)]


def load_specs(path: str) -> List[Spec]:
    with open(path) as f:
        raw = yaml.safe_load(f) or []
    if not isinstance(raw, list):
        raise ValueError(
            f"plugin specs file must be a YAML list, got {type(raw).__name__}"
        )
    for i, d in enumerate(raw):
        if not isinstance(d, dict):
            raise ValueError(f"plugin spec #{i} is not a mapping")
    specs = [Spec.from_dict(d) for d in raw]
    names = set()
    for s in specs:
        err = s.validate()
        if err:
            raise ValueError(f"plugin {s.plugin_name!r}: {err}")
        if s.plugin_name in names:
            raise ValueError(f"duplicate plugin name {s.plugin_name!r}")
        names.add(s.plugin_name)
    return specs


def _dig(obj: Any, dotpath: str) -> Any:
    cur = obj
    for part in dotpath.lstrip(".").split("."):
        if not part:
            continue
        if isinstance(cur, dict) and part in cur:
            cur = cur[part]
        else:
            return None
    return cur


class PluginComponent(TickerComponent):
    """One custom plugin as a registry component."""

    def __init__(
        self,
        spec: Spec,
        param: str = "",
        entry_name: str = "",
        run_mode_override: str = "",
    ):
        super().__init__()
        self.spec = spec
        self.param = param
        self.entry_name = entry_name
        self.poll_interval = spec.interval_seconds
        # component-specific run_mode wins over the parent plugin's
        # (docs/PLUGIN.md "Parameter Inheritance and Priority")
        self.run_mode = run_mode_override or spec.run_mode

    @property
    def name(self) -> str:
        sub = self.entry_name or self.param
        if sub:
            return f"{self.spec.component_name}-{sub}"
        return self.spec.component_name

    def tags(self) -> List[str]:
        return list(self.spec.tags) or [self.name]

    def deregisterable(self) -> bool:
        return True  # custom plugins can be deregistered (reference behavior)

    def check(self) -> CheckResult:
        outputs = []
        for step in self.spec.steps:
            script = step.decoded()
            # ${NAME} = component entry name, ${PAR} = its parameter
            # (docs/PLUGIN.md "Parameter Substitution"); name-only entries
            # historically doubled as the parameter, so ${NAME} falls back
            # to the param when no entry name was parsed
            if self.entry_name or self.param:
                script = script.replace("${NAME}", self.entry_name or self.param)
                script = script.replace("${PAR}", self.param)
            res = run_bash(script, timeout_seconds=self.spec.timeout_seconds)
            outputs.append(res.output)
            if res.timed_out or res.exit_code != 0:
                return CheckResult(
                    self.name,
                    health=HealthStateType.UNHEALTHY,
                    reason=(
                        "step timed out"
                        if res.timed_out
                        else f"step exited {res.exit_code}"
                    ),
                    raw_output="\n".join(outputs)[-4096:],
                    component_type=ComponentType.CUSTOM_PLUGIN,
                    run_mode=self.spec.run_mode,
                )
        raw = "\n".join(outputs)
        self._log_output(raw)
        health, reason, actions, extra = self._parse(raw)
        return CheckResult(
            self.name,
            health=health,
            reason=reason,
            raw_output=raw[-4096:],
            extra_info=extra or None,
            suggested_actions=actions,
            component_type=ComponentType.CUSTOM_PLUGIN,
            run_mode=self.spec.run_mode,
        )

    def _log_output(self, raw: str) -> None:
        """Append one RFC3339-stamped line to the parser's log_path with
        ${PLUGIN}/${TRIGGER} substituted; skipped when a referenced
        variable is empty (docs/PLUGIN.md Log Path Variable Substitution)."""
        path = self.spec.log_path
        if not path:
            return
        plugin, trigger = self.spec.plugin_name, (self.run_mode or "")
        if "${PLUGIN}" in path and not plugin:
            return
        if "${TRIGGER}" in path and not trigger:
            return
        path = path.replace("${PLUGIN}", plugin).replace("${TRIGGER}", trigger)
        try:
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            from ..apiv1.types import rfc3339, utcnow

            with open(path, "a") as f:
                f.write(
                    f"[{rfc3339(utcnow())}] plugin={plugin} "
                    f"trigger={trigger} output={raw.strip()[-2048:]}\n"
                )
        except OSError:
            pass  # logging is best-effort; never fail the check over it

    def _parse(self, raw: str):
        if not self.spec.json_paths:
            return HealthStateType.HEALTHY, "plugin succeeded", None, {}
        # the reference extracts the FIRST valid JSON object found in the
        # output, even when embedded in other text (docs/PLUGIN.md parser
        # notes); line-leading objects are tried first, then an embedded
        # brace-scan, then the legacy last-line pass
        parsed = None
        for line in raw.strip().splitlines():
            line = line.strip()
            if line.startswith("{"):
                try:
                    parsed = json.loads(line)
                    break
                except json.JSONDecodeError:
                    continue
        if parsed is None:
            decoder = json.JSONDecoder()
            idx = raw.find("{")
            while idx != -1 and parsed is None:
                try:
                    candidate, _end = decoder.raw_decode(raw[idx:])
                    if isinstance(candidate, dict):
                        parsed = candidate
                except json.JSONDecodeError:
                    pass
                idx = raw.find("{", idx + 1)
        if parsed is None:
            return (
                HealthStateType.UNHEALTHY,
                "plugin output has no parsable JSON line",
                None,
                {},
            )
        extra: Dict[str, str] = {}
        actions: List[str] = []
        for rule in self.spec.json_paths:
            val = _dig(parsed, rule.query)
            sval = "" if val is None else str(val)
            if rule.field:
                extra[rule.field] = sval
            if rule.expect_regex and not re.search(rule.expect_regex, sval):
                return (
                    HealthStateType.UNHEALTHY,
                    f"field {rule.query!r} value {sval!r} does not match "
                    f"{rule.expect_regex!r}",
                    None,
                    extra,
                )
            for action, regex in rule.suggested_actions.items():
                if re.search(regex, sval) and action not in actions:
                    actions.append(action)
        sa = (
            SuggestedActions(description="plugin-suggested", repair_actions=actions)
            if actions
            else None
        )
        health = HealthStateType.HEALTHY
        return health, "plugin succeeded", sa, extra


def parse_component_list_entry(entry: str):
    """One component-list entry -> (name, run_mode_override, param).

    Reference formats (docs/PLUGIN.md "Component List Format"):
    ``name#run_mode:param`` | ``name#run_mode`` | ``name:param`` | ``name``.
    run_mode has the highest priority for that component; timeout/interval
    always inherit from the parent plugin."""
    run_mode = ""
    param = ""
    name = entry
    if "#" in entry:
        head, rest = entry.split("#", 1)
        mode = rest.split(":", 1)[0]
        # '#' only separates a run_mode; otherwise it is part of the name
        # (the doc's indent-escape for names starting with '#')
        if mode in ("auto", "manual", "once"):
            name = head
            if ":" in rest:
                run_mode, param = rest.split(":", 1)
            else:
                run_mode = rest
        elif ":" in entry:
            name, param = entry.split(":", 1)
    elif ":" in entry:
        name, param = entry.split(":", 1)
    return name.strip(), run_mode.strip(), param


def load_component_list_file(path: str) -> List[str]:
    """Plain-text component list: one entry per line; empty lines and lines
    whose FIRST character is ``#`` are ignored (a name starting with # can
    be escaped by indenting with a space — docs/PLUGIN.md:174)."""
    entries = []
    with open(path) as f:
        for line in f:
            line = line.rstrip("\n")
            if not line.strip() or line.startswith("#"):
                continue
            entries.append(line.strip())
    return entries


def make_components(spec: Spec) -> List[Component]:
    if spec.plugin_type == PLUGIN_TYPE_COMPONENT_LIST:
        entries = list(spec.component_list)
        if spec.component_list_file:
            entries.extend(load_component_list_file(spec.component_list_file))
        out = []
        for entry in entries:
            name, run_mode, param = parse_component_list_entry(entry)
            out.append(
                PluginComponent(
                    spec, param=param, entry_name=name, run_mode_override=run_mode
                )
            )
        return out
    return [PluginComponent(spec)]


def run_init_plugins(specs: List[Spec]) -> Optional[str]:
    """Run init-type plugins once; returns an error string on failure
    (reference: init plugins gate server start, server.go:377-389)."""
    for spec in specs:
        if spec.plugin_type != PLUGIN_TYPE_INIT:
            continue
        comp = PluginComponent(spec)
        cr = comp.trigger_check()
        if cr.health != HealthStateType.HEALTHY:
            return f"init plugin {spec.plugin_name!r} failed: {cr.reason}"
    return None
