"""helmlite: a minimal Helm-template renderer for this chart's subset.

There is no network (and no ``helm`` binary) in the CI image, but the chart
must still be rendered and structurally verified — the check the reference
gets from ``helm template`` in its CI (``tests/bats/test_basics.bats``).
helmlite implements exactly the Go-template subset our chart uses, so CI
renders the same manifests a real ``helm template`` would:

* ``{{ define "name" }}…{{ end }}`` blocks (from ``_helpers.tpl``),
* ``{{ include "name" . }}`` (root context only),
* ``.Values.*`` / ``.Release.Name`` / ``.Release.Namespace`` /
  ``.Chart.Name`` / ``.Chart.Version`` / ``.Chart.AppVersion`` lookups,
* ``range $x := <list>`` / ``end`` over ``splitList`` results, ``toYaml``,
* ``if`` / ``else`` / ``end`` with Helm truthiness (empty string, false,
  nil, 0 are falsy) and the ``eq``, ``ne``, ``and``, ``or``, ``not``
  functions over literals and lookups,
* pipelines: ``quote``, ``trim``, ``nindent N``, ``indent N``,
  ``trunc N``, ``trimSuffix S``, ``default D``, ``lower``, ``replace A B``,
* whitespace-chomping ``{{-`` and ``-}}``.

Anything outside the subset raises, so chart drift into unsupported
constructs fails CI instead of silently rendering wrong.
"""

from __future__ import annotations

import os
import re
from typing import Any, Dict, List, Optional, Tuple

__all__ = ["render_chart", "render_template", "HelmliteError"]


class HelmliteError(ValueError):
    pass


_MISSING = object()


_ACTION_RE = re.compile(r"\{\{-?\s*(.*?)\s*-?\}\}", re.DOTALL)


def _chomp(src: str) -> str:
    """Apply {{- and -}} whitespace chomping by rewriting the surrounding
    text before tokenization."""
    src = re.sub(r"[ \t]*\n?[ \t]*\{\{-", "{{", src)
    src = re.sub(r"-\}\}[ \t]*\n?", "}}", src)
    return src


def _tokenize(src: str) -> List[Tuple[str, str]]:
    """-> [(kind, payload)]: kind in {text, action}."""
    out = []
    pos = 0
    src = _chomp(src)
    for m in _ACTION_RE.finditer(src):
        if m.start() > pos:
            out.append(("text", src[pos:m.start()]))
        out.append(("action", m.group(1).strip()))
        pos = m.end()
    if pos < len(src):
        out.append(("text", src[pos:]))
    return out


def _truthy(v: Any) -> bool:
    if v is None or v is False:
        return False
    if isinstance(v, (str, list, dict)) and len(v) == 0:
        return False
    if isinstance(v, (int, float)) and v == 0:
        return False
    return True


class _Renderer:
    def __init__(self, values: Dict[str, Any], chart_meta: Dict[str, Any],
                 release: Dict[str, str], defines: Dict[str, List[Tuple[str, str]]]):
        self.values = values
        self.chart = chart_meta
        self.release = release
        self.defines = defines
        self.vars: Dict[str, Any] = {}  # $name bindings from range

    # -- expression evaluation ------------------------------------------

    def _lookup(self, dotted: str) -> Any:
        if dotted in (".", "$"):
            return self  # context marker; only used as include arg
        parts = dotted.lstrip(".").split(".")
        if not parts or not parts[0]:
            raise HelmliteError(f"bad lookup {dotted!r}")
        head = parts[0]
        if head == "Values":
            cur: Any = self.values
            walk = parts[1:]
        elif head == "Release":
            cur = self.release
            walk = parts[1:]
        elif head == "Chart":
            cur = self.chart
            walk = parts[1:]
        else:
            raise HelmliteError(f"unsupported root {dotted!r}")
        for p in walk:
            if isinstance(cur, dict) and p in cur:
                cur = cur[p]
            else:
                return None
        return cur

    def _split_args(self, s: str) -> List[str]:
        """Split on spaces outside quotes and parens."""
        args, buf, depth, q = [], "", 0, None
        for ch in s:
            if q:
                buf += ch
                if ch == q:
                    q = None
                continue
            if ch in "\"'":
                q = ch
                buf += ch
            elif ch == "(":
                depth += 1
                buf += ch
            elif ch == ")":
                depth -= 1
                buf += ch
            elif ch == " " and depth == 0:
                if buf:
                    args.append(buf)
                    buf = ""
            else:
                buf += ch
        if buf:
            args.append(buf)
        return args

    def _eval_term(self, term: str) -> Any:
        term = term.strip()
        if term.startswith("(") and term.endswith(")"):
            return self._eval_expr(term[1:-1])
        if term.startswith('"') and term.endswith('"'):
            return term[1:-1]
        if term.startswith("'") and term.endswith("'"):
            return term[1:-1]
        if re.fullmatch(r"-?\d+", term):
            return int(term)
        if term in ("true", "false"):
            return term == "true"
        if term.startswith(".") or term in (".", "$"):
            return self._lookup(term)
        if term.startswith("$"):
            name = term[1:]
            if name not in self.vars:
                raise HelmliteError(f"undefined variable {term!r}")
            return self.vars[name]
        raise HelmliteError(f"unsupported term {term!r}")

    def _eval_expr(self, expr: str) -> Any:
        expr = expr.strip()
        args = self._split_args(expr)
        if not args:
            raise HelmliteError("empty expression")
        head = args[0]
        if head == "include":
            if len(args) != 3:
                raise HelmliteError(f"include wants 2 args: {expr!r}")
            name = self._eval_term(args[1])
            if name not in self.defines:
                raise HelmliteError(f"include of undefined template {name!r}")
            return self._render_tokens(self.defines[name]).strip("\n")
        if head == "eq":
            return self._eval_term(args[1]) == self._eval_term(args[2])
        if head == "ne":
            return self._eval_term(args[1]) != self._eval_term(args[2])
        if head == "and":
            v: Any = True
            for a in args[1:]:
                v = self._eval_term(a)
                if not _truthy(v):
                    return v
            return v
        if head == "or":
            for a in args[1:]:
                v = self._eval_term(a)
                if _truthy(v):
                    return v
            return v
        if head == "not":
            return not _truthy(self._eval_term(args[1]))
        if head == "default":
            # prefix form: default DEFAULT VALUE
            v = self._eval_term(args[2])
            return v if _truthy(v) else self._eval_term(args[1])
        if head == "fail":
            raise HelmliteError(f"chart validation failed: {self._eval_term(args[1])}")
        if head == "splitList":
            sep = str(self._eval_term(args[1]))
            return str(self._eval_term(args[2]) or "").split(sep)
        if head == "printf":
            fmt = self._eval_term(args[1])
            vals = tuple(self._eval_term(a) for a in args[2:])
            return fmt.replace("%s", "{}").replace("%d", "{}").format(*vals)
        if len(args) == 1:
            return self._eval_term(head)
        raise HelmliteError(f"unsupported function {head!r} in {expr!r}")

    def _eval_pipeline(self, action: str) -> Any:
        stages = [s.strip() for s in self._split_pipes(action)]
        val = self._eval_expr(stages[0])
        for st in stages[1:]:
            parts = self._split_args(st)
            fn, fargs = parts[0], [self._eval_term(a) for a in parts[1:]]
            if fn == "quote":
                val = '"%s"' % str("" if val is None else val)
            elif fn == "trim":
                val = str(val or "").strip()
            elif fn == "lower":
                val = str(val or "").lower()
            elif fn == "trunc":
                val = str(val or "")[: int(fargs[0])]
            elif fn == "trimSuffix":
                s = str(val or "")
                suf = str(fargs[0])
                val = s[: -len(suf)] if suf and s.endswith(suf) else s
            elif fn == "default":
                val = val if _truthy(val) else fargs[0]
            elif fn == "replace":
                val = str(val or "").replace(str(fargs[0]), str(fargs[1]))
            elif fn == "indent":
                pad = " " * int(fargs[0])
                val = "\n".join(pad + l for l in str(val or "").splitlines())
            elif fn == "toYaml":
                import yaml as _yaml
                val = _yaml.safe_dump(val, default_flow_style=False,
                                      sort_keys=False).rstrip("\n")
            elif fn == "nindent":
                pad = " " * int(fargs[0])
                val = "\n" + "\n".join(pad + l for l in str(val or "").splitlines())
            else:
                raise HelmliteError(f"unsupported pipe function {fn!r}")
        return val

    def _split_pipes(self, s: str) -> List[str]:
        out, buf, depth, q = [], "", 0, None
        for ch in s:
            if q:
                buf += ch
                if ch == q:
                    q = None
                continue
            if ch in "\"'":
                q = ch
                buf += ch
            elif ch == "(":
                depth += 1
                buf += ch
            elif ch == ")":
                depth -= 1
                buf += ch
            elif ch == "|" and depth == 0:
                out.append(buf)
                buf = ""
            else:
                buf += ch
        out.append(buf)
        return out

    # -- block structure -------------------------------------------------

    def _render_tokens(self, tokens: List[Tuple[str, str]]) -> str:
        out, _ = self._render_block(tokens, 0, None)
        return out

    def _render_block(self, tokens, i, until) -> Tuple[str, int]:
        """Render until an 'end'/'else' terminator (when until='if')."""
        out: List[str] = []
        while i < len(tokens):
            kind, payload = tokens[i]
            if kind == "text":
                out.append(payload)
                i += 1
                continue
            if payload.startswith("/*"):  # {{/* comment */}}
                i += 1
                continue
            word = payload.split(None, 1)[0] if payload else ""
            if word == "if":
                cond = self._eval_pipeline(payload[2:].strip())
                body, i = self._collect_if(tokens, i + 1)
                chosen = body["then"] if _truthy(cond) else body["else"]
                rendered, _ = _Renderer._render_block(self, chosen, 0, None)
                out.append(rendered)
                continue
            if word == "range":
                # range $var := <pipeline>
                m = re.fullmatch(r"range\s+\$(\w+)\s*:=\s*(.+)", payload, re.DOTALL)
                if not m:
                    raise HelmliteError(f"unsupported range form {payload!r}")
                var, expr = m.group(1), m.group(2)
                items = self._eval_pipeline(expr)
                if items is None:
                    items = []
                if not isinstance(items, (list, tuple)):
                    raise HelmliteError(f"range needs a list, got {type(items).__name__}")
                body, i = self._collect_body(tokens, i + 1)
                saved = self.vars.get(var, _MISSING)
                for item in items:
                    self.vars[var] = item
                    rendered, _ = self._render_block(body, 0, None)
                    out.append(rendered)
                if saved is _MISSING:
                    self.vars.pop(var, None)
                else:
                    self.vars[var] = saved
                continue
            if word in ("end", "else"):
                if until is None:
                    raise HelmliteError(f"unexpected {word!r}")
                return "".join(out), i
            if word == "define":
                # defines are pre-extracted; skip blocks when encountered
                _, i = self._skip_to_end(tokens, i + 1)
                continue
            val = self._eval_pipeline(payload)
            out.append(str("" if val is None else val))
            i += 1
        if until is not None:
            raise HelmliteError("unterminated block")
        return "".join(out), i

    def _collect_if(self, tokens, i):
        """Collect then/else token lists of an if-block (supports else if)."""
        then: List = []
        els: List = []
        cur = then
        depth = 0
        while i < len(tokens):
            kind, payload = tokens[i]
            word = payload.split(None, 1)[0] if kind == "action" and payload else ""
            if kind == "action" and word in ("if", "define", "range", "with"):
                depth += 1
            elif kind == "action" and word == "end":
                if depth == 0:
                    return {"then": then, "else": els}, i + 1
                depth -= 1
            elif kind == "action" and word == "else" and depth == 0:
                rest = payload[4:].strip()
                if rest.startswith("if"):
                    # else-if: the rest becomes a nested if inside else
                    els = [("action", rest)]
                    cur = els
                    # collect the remainder into the nested if's scope by
                    # continuing; the nested if consumes the shared 'end',
                    # so bump depth bookkeeping via recursion instead:
                    i += 1
                    tail: List = []
                    d2 = 0
                    while i < len(tokens):
                        k2, p2 = tokens[i]
                        w2 = p2.split(None, 1)[0] if k2 == "action" and p2 else ""
                        if k2 == "action" and w2 in ("if", "define", "range", "with"):
                            d2 += 1
                        elif k2 == "action" and w2 == "end":
                            if d2 == 0:
                                els.extend(tail)
                                els.append(("action", "end"))
                                return {"then": then, "else": els}, i + 1
                            d2 -= 1
                        tail.append(tokens[i])
                        i += 1
                    raise HelmliteError("unterminated else-if")
                cur = els
                i += 1
                continue
            cur.append(tokens[i])
            i += 1
        raise HelmliteError("unterminated if")

    def _collect_body(self, tokens, i):
        """Collect a range body up to its matching end."""
        body: List = []
        depth = 0
        while i < len(tokens):
            kind, payload = tokens[i]
            word = payload.split(None, 1)[0] if kind == "action" and payload else ""
            if kind == "action" and word in ("if", "define", "range", "with"):
                depth += 1
            elif kind == "action" and word == "end":
                if depth == 0:
                    return body, i + 1
                depth -= 1
            body.append(tokens[i])
            i += 1
        raise HelmliteError("unterminated range")

    def _skip_to_end(self, tokens, i):
        depth = 0
        while i < len(tokens):
            kind, payload = tokens[i]
            word = payload.split(None, 1)[0] if kind == "action" and payload else ""
            if kind == "action" and word in ("if", "define", "range", "with"):
                depth += 1
            elif kind == "action" and word == "end":
                if depth == 0:
                    return None, i + 1
                depth -= 1
            i += 1
        raise HelmliteError("unterminated define")


def _extract_defines(tokens) -> Dict[str, List[Tuple[str, str]]]:
    defines: Dict[str, List[Tuple[str, str]]] = {}
    i = 0
    while i < len(tokens):
        kind, payload = tokens[i]
        if kind == "action" and payload.startswith("define"):
            m = re.match(r'define\s+"([^"]+)"', payload)
            if not m:
                raise HelmliteError(f"bad define {payload!r}")
            name = m.group(1)
            body: List[Tuple[str, str]] = []
            depth = 0
            i += 1
            while i < len(tokens):
                k2, p2 = tokens[i]
                w2 = p2.split(None, 1)[0] if k2 == "action" and p2 else ""
                if k2 == "action" and w2 in ("if", "define", "range", "with"):
                    depth += 1
                elif k2 == "action" and w2 == "end":
                    if depth == 0:
                        break
                    depth -= 1
                body.append(tokens[i])
                i += 1
            defines[name] = body
        i += 1
    return defines


def _deep_merge(base: Dict, override: Dict) -> Dict:
    out = dict(base)
    for k, v in (override or {}).items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = v
    return out


def render_chart(chart_dir: str, values_override: Optional[Dict[str, Any]] = None,
                 release_name: str = "amd-dra-driver",
                 namespace: str = "amd-dra-driver") -> Dict[str, str]:
    """Render every template in `chart_dir` -> {filename: rendered_yaml}."""
    import yaml

    with open(os.path.join(chart_dir, "values.yaml")) as f:
        values = yaml.safe_load(f) or {}
    values = _deep_merge(values, values_override or {})
    with open(os.path.join(chart_dir, "Chart.yaml")) as f:
        chart_yaml = yaml.safe_load(f) or {}
    chart_meta = {
        "Name": chart_yaml.get("name", ""),
        "Version": chart_yaml.get("version", ""),
        "AppVersion": chart_yaml.get("appVersion", ""),
    }
    release = {"Name": release_name, "Namespace": namespace, "Service": "Helm"}

    tdir = os.path.join(chart_dir, "templates")
    defines: Dict[str, List[Tuple[str, str]]] = {}
    helpers = os.path.join(tdir, "_helpers.tpl")
    if os.path.exists(helpers):
        defines.update(_extract_defines(_tokenize(open(helpers).read())))

    out: Dict[str, str] = {}
    for fname in sorted(os.listdir(tdir)):
        if not fname.endswith(".yaml"):
            continue
        tokens = _tokenize(open(os.path.join(tdir, fname)).read())
        defines.update(_extract_defines(tokens))
        r = _Renderer(values, chart_meta, release, defines)
        out[fname] = r._render_tokens(tokens)
    return out


def chart_deviceclasses(chart_dir: str,
                        resource_api_version: str = "v1") -> List[Dict[str, Any]]:
    """Render the chart's DeviceClass manifests (default: the v1 surface a
    k8s >= 1.35 cluster serves, with extendedResourceName on gpu.amd.com).
    Shared by the e2e tests, bench localcluster and soak harnesses."""
    import yaml

    rendered = render_chart(chart_dir, {"resourceApiVersion": resource_api_version})
    return [d for d in yaml.safe_load_all(rendered["deviceclasses.yaml"]) if d]


def render_template(src: str, values: Dict[str, Any],
                    defines_src: str = "",
                    release: Optional[Dict[str, str]] = None,
                    chart_meta: Optional[Dict[str, Any]] = None) -> str:
    tokens = _tokenize(src)
    defines = _extract_defines(_tokenize(defines_src)) if defines_src else {}
    defines.update(_extract_defines(tokens))
    r = _Renderer(values, chart_meta or {"Name": "t", "Version": "0", "AppVersion": "0"},
                  release or {"Name": "test", "Namespace": "default", "Service": "Helm"},
                  defines)
    return r._render_tokens(tokens)
