"""helmlite: a renderer for the Go-template subset used by this repo's Helm
chart, plus a helm-unittest-spec runner.

Why it exists: the build/CI image has no `helm` binary, so the chart's
helm-unittest specs (helm/tests/*_test.yaml — same format as the reference's
24 specs under reference helm/tests/) would otherwise be dead weight. This
module renders the chart's templates faithfully enough to execute those
specs as ordinary pytest tests on CPU.

Supported template language (everything the chart uses):
  {{ .Values.a.b }}  {{- ... -}} chomping   {{ $var := expr }}  {{ $ }}
  if / else if / else / end, range ($k, $v :=), with, define/include
  pipelines with: default quote squote indent nindent toYaml toJson
  eq ne not and or dict list until add len printf b64enc lower upper
  .Release.Name/.Namespace/.Service, .Chart.Name/.Version,
  .Files.Get, .Capabilities.KubeVersion
"""

from __future__ import annotations

import base64
import json
import os
import re
from typing import Any, Dict, List, Optional, Tuple

import yaml

CHART_DIR = os.path.join(os.path.dirname(__file__), "..", "helm")

_ACTION = re.compile(r"\{\{-?\s*(.*?)\s*-?\}\}", re.S)
_TOKEN = re.compile(
    r"\"(?:[^\"\\]|\\.)*\"|'(?:[^'\\]|\\.)*'|\(|\)|\||:=|[^\s()|]+"
)


def _chomp(src: str) -> List[Tuple[str, str]]:
    """Split template into [('text', s) | ('action', code)] applying the
    {{- / -}} whitespace chomping rules."""
    parts: List[Tuple[str, str]] = []
    pos = 0
    for m in re.finditer(r"\{\{(-?)\s*(.*?)\s*(-?)\}\}", src, re.S):
        text = src[pos:m.start()]
        if m.group(1) == "-":
            text = text.rstrip()
        parts.append(("text", text))
        parts.append(("action", m.group(2)))
        pos = m.end()
        if m.group(3) == "-":
            rest = src[pos:]
            stripped = rest.lstrip()
            pos += len(rest) - len(stripped)
    parts.append(("text", src[pos:]))
    return parts


# ---- expression evaluation -------------------------------------------------
class _Undefined:
    def __repr__(self) -> str:
        return "<no value>"


UNDEF = _Undefined()


def _truthy(v: Any) -> bool:
    if v is UNDEF or v is None:
        return False
    if isinstance(v, (list, dict, str)):
        return len(v) > 0
    return bool(v)


def _to_yaml(v: Any) -> str:
    if v is UNDEF or v is None:
        return ""
    return yaml.safe_dump(v, default_flow_style=False, sort_keys=False).rstrip(
        "\n"
    )


def _indent(n: int, s: Any) -> str:
    pad = " " * int(n)
    return "\n".join(pad + line for line in _render_scalar(s).split("\n"))


def _render_scalar(v: Any) -> str:
    if v is UNDEF or v is None:
        return ""
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, float) and v == int(v):
        return str(int(v))
    return str(v)


class Renderer:
    def __init__(self, values: Dict[str, Any], release: str = "release",
                 namespace: str = "default") -> None:
        self.defines: Dict[str, list] = {}
        self.root = {
            "Values": values,
            "Release": {"Name": release, "Namespace": namespace,
                        "Service": "Helm"},
            "Chart": {"Name": "production-stack-amd", "Version": "0.1.0"},
            "Capabilities": {"KubeVersion": {"Version": "v1.30.0"}},
            "Files": _Files(),
        }

    # -- expression parser ----------------------------------------------
    def eval_expr(self, code: str, dot: Any, vars: Dict[str, Any]) -> Any:
        toks = _TOKEN.findall(code)
        segments: List[List[str]] = [[]]
        depth = 0
        for t in toks:
            if t == "(":
                depth += 1
                segments[-1].append(t)
            elif t == ")":
                depth -= 1
                segments[-1].append(t)
            elif t == "|" and depth == 0:
                segments.append([])
            else:
                segments[-1].append(t)
        val = self._eval_call(segments[0], dot, vars)
        for seg in segments[1:]:
            val = self._eval_call(seg, dot, vars, piped=val)
        return val

    def _eval_atom(self, tok: str, dot: Any, vars: Dict[str, Any]) -> Any:
        if tok.startswith('"') or tok.startswith("'"):
            body = tok[1:-1]
            return body.replace('\\"', '"').replace("\\n", "\n")
        if re.fullmatch(r"-?\d+", tok):
            return int(tok)
        if re.fullmatch(r"-?\d+\.\d+", tok):
            return float(tok)
        if tok == "true":
            return True
        if tok == "false":
            return False
        if tok in ("nil", "null"):
            return None
        if tok == "$":
            return self.root
        if tok == ".":
            return dot
        if tok.startswith("$"):
            path = tok[1:].split(".")
            base = self.root if path[0] == "" else vars.get(path[0], UNDEF)
            return self._walk(base, path[1:])
        if tok.startswith("."):
            parts = [p for p in tok.split(".") if p]
            if parts and parts[0] in ("Values", "Release", "Chart", "Files",
                                      "Capabilities"):
                return self._walk(self.root, parts)
            return self._walk(dot, parts)
        raise ValueError(f"unknown token {tok!r}")

    def _walk(self, base: Any, parts: List[str]) -> Any:
        cur = base
        for p in parts:
            if cur is UNDEF or cur is None:
                return UNDEF
            if isinstance(cur, dict):
                cur = cur.get(p, UNDEF)
            else:
                cur = getattr(cur, p, UNDEF)
        return cur

    _NO_PIPE = object()

    def _eval_call(self, toks: List[str], dot: Any, vars: Dict[str, Any],
                   piped: Any = _NO_PIPE) -> Any:
        # resolve parenthesized sub-expressions first
        resolved: List[Any] = []
        i = 0
        while i < len(toks):
            if toks[i] == "(":
                depth = 1
                j = i + 1
                while j < len(toks) and depth:
                    if toks[j] == "(":
                        depth += 1
                    elif toks[j] == ")":
                        depth -= 1
                    j += 1
                inner = " ".join(toks[i + 1:j - 1])
                resolved.append(self.eval_expr(inner, dot, vars))
                i = j
            else:
                resolved.append(toks[i])
                i += 1
        if not resolved:
            return UNDEF if piped is self._NO_PIPE else piped
        head = resolved[0]
        args = resolved[1:]

        def ev(x: Any) -> Any:
            if isinstance(x, str) and x in ("dict", "list"):
                return {} if x == "dict" else []
            return self._eval_atom(x, dot, vars) if isinstance(x, str) else x

        if isinstance(head, str) and head in _FUNCS:
            fargs = [ev(a) for a in args]
            # helm semantics: a nil/missing PIPED value still reaches the
            # function (quote -> "", toYaml -> "", default -> fallback);
            # a call with no pipe at all gets only its literal args
            if piped is not self._NO_PIPE:
                fargs.append(piped)
            return _FUNCS[head](self, dot, vars, *fargs)
        if args:  # e.g. `.Files.Get "path"` method call style
            base = ev(head)
            if callable(base):
                return base(*[ev(a) for a in args])
            raise ValueError(f"cannot call {head!r}")
        return ev(head)

    # -- block parser ----------------------------------------------------
    def parse(self, src: str) -> list:
        parts = _chomp(src)
        nodes, _ = self._parse_block(parts, 0, None)
        return nodes

    def _parse_block(self, parts, i, until) -> Tuple[list, int]:
        nodes: list = []
        while i < len(parts):
            kind, body = parts[i]
            if kind == "text":
                nodes.append(("text", body))
                i += 1
                continue
            word = body.split(None, 1)[0] if body else ""
            if word in ("end", "else") and until:
                return nodes, i
            if word == "if":
                cond = body[2:].strip()
                then, i = self._parse_block(parts, i + 1, "ifelse")
                clauses = [(cond, then)]
                els: list = []
                while parts[i][1].startswith("else"):
                    rest = parts[i][1][4:].strip()
                    if rest.startswith("if"):
                        sub, i = self._parse_block(parts, i + 1, "ifelse")
                        clauses.append((rest[2:].strip(), sub))
                    else:
                        els, i = self._parse_block(parts, i + 1, "ifelse")
                        break
                assert parts[i][1].split(None, 1)[0] == "end"
                nodes.append(("if", clauses, els))
                i += 1
            elif word == "range":
                expr = body[5:].strip()
                inner, i = self._parse_block(parts, i + 1, "loop")
                els = []
                if parts[i][1].startswith("else"):
                    els, i = self._parse_block(parts, i + 1, "loop")
                nodes.append(("range", expr, inner, els))
                i += 1
            elif word == "with":
                expr = body[4:].strip()
                inner, i = self._parse_block(parts, i + 1, "loop")
                els = []
                if parts[i][1].startswith("else"):
                    els, i = self._parse_block(parts, i + 1, "loop")
                nodes.append(("with", expr, inner, els))
                i += 1
            elif word == "define":
                name = body[6:].strip().strip('"')
                inner, i = self._parse_block(parts, i + 1, "loop")
                self.defines[name] = inner
                i += 1
            elif word.startswith("/*") or body.startswith("/*"):
                i += 1  # comment
            else:
                nodes.append(("action", body))
                i += 1
        return nodes, i

    # -- execution -------------------------------------------------------
    def exec_nodes(self, nodes: list, dot: Any, vars: Dict[str, Any],
                   out: List[str]) -> None:
        for node in nodes:
            tag = node[0]
            if tag == "text":
                out.append(node[1])
            elif tag == "action":
                body = node[1]
                m = re.match(r"(\$\w+)\s*:?=\s*(.*)", body, re.S)
                if m:
                    vars[m.group(1)[1:]] = self.eval_expr(
                        m.group(2), dot, vars)
                    continue
                val = self.eval_expr(body, dot, vars)
                out.append(_render_scalar(val))
            elif tag == "if":
                _, clauses, els = node
                done = False
                for cond, sub in clauses:
                    if _truthy(self.eval_expr(cond, dot, vars)):
                        self.exec_nodes(sub, dot, vars, out)
                        done = True
                        break
                if not done:
                    self.exec_nodes(els, dot, vars, out)
            elif tag == "range":
                _, expr, inner, els = node
                m = re.match(r"(\$\w+)\s*,\s*(\$\w+)\s*:=\s*(.*)", expr)
                m1 = re.match(r"(\$\w+)\s*:=\s*(.*)", expr) if not m else None
                coll = self.eval_expr(
                    m.group(3) if m else (m1.group(2) if m1 else expr),
                    dot, vars)
                items: List[Tuple[Any, Any]] = []
                if isinstance(coll, list):
                    items = list(enumerate(coll))
                elif isinstance(coll, dict):
                    items = list(coll.items())
                if not items:
                    self.exec_nodes(els, dot, vars, out)
                    continue
                # Go-template `$x = y` (reassignment) must be visible in
                # the enclosing scope (the chart's $hasSecrets idiom), so
                # the loop shares `vars` and only the loop variables are
                # restored afterwards.
                names = []
                if m:
                    names = [m.group(1)[1:], m.group(2)[1:]]
                elif m1:
                    names = [m1.group(1)[1:]]
                saved = {n: vars.get(n, UNDEF) for n in names}
                for k, v in items:
                    if m:
                        vars[names[0]] = k
                        vars[names[1]] = v
                    elif m1:
                        vars[names[0]] = v
                    self.exec_nodes(inner, v, vars, out)
                for n, old in saved.items():
                    if old is UNDEF:
                        vars.pop(n, None)
                    else:
                        vars[n] = old
            elif tag == "with":
                _, expr, inner, els = node
                val = self.eval_expr(expr, dot, vars)
                if _truthy(val):
                    self.exec_nodes(inner, val, vars, out)
                else:
                    self.exec_nodes(els, dot, vars, out)

    def render(self, src: str) -> str:
        nodes = self.parse(src)
        out: List[str] = []
        self.exec_nodes(nodes, self.root, {}, out)
        return "".join(out)

    def include(self, name: str, ctx: Any) -> str:
        out: List[str] = []
        self.exec_nodes(self.defines.get(name, []), ctx, {}, out)
        return "".join(out)


class _Files:
    def Get(self, path: str) -> str:
        p = os.path.join(CHART_DIR, path)
        with open(p) as f:
            return f.read()


def _f_default(r, dot, vars, *a):
    # helm: `x | default y` -> default(y_default=a[0], piped=a[1])
    dflt, val = a[0], a[-1]
    return dflt if (val is UNDEF or val is None or val == "" or val == 0 or
                    val is False) else val


_FUNCS = {
    "default": _f_default,
    "quote": lambda r, d, v, x: '"' + _render_scalar(x) + '"',
    "squote": lambda r, d, v, x: "'" + _render_scalar(x) + "'",
    "indent": lambda r, d, v, n, x: _indent(n, x),
    "nindent": lambda r, d, v, n, x: "\n" + _indent(n, x),
    "toYaml": lambda r, d, v, x: _to_yaml(x),
    "toJson": lambda r, d, v, x: json.dumps(
        None if x is UNDEF else x),
    "b64enc": lambda r, d, v, x: base64.b64encode(
        _render_scalar(x).encode()).decode(),
    "lower": lambda r, d, v, x: _render_scalar(x).lower(),
    "upper": lambda r, d, v, x: _render_scalar(x).upper(),
    "eq": lambda r, d, v, a, b: a == b,
    "ne": lambda r, d, v, a, b: a != b,
    "not": lambda r, d, v, x: not _truthy(x),
    "and": lambda r, d, v, *a: all(_truthy(x) for x in a),
    "or": lambda r, d, v, *a: next((x for x in a if _truthy(x)), a[-1]),
    "add": lambda r, d, v, *a: sum(a),
    "sub": lambda r, d, v, a, b: a - b,
    "len": lambda r, d, v, x: len(x) if x is not UNDEF else 0,
    "printf": lambda r, d, v, fmt, *a: _go_printf(fmt, a),
    "list": lambda r, d, v, *a: list(a),
    "until": lambda r, d, v, n: list(range(int(n))),
    "dict": lambda r, d, v, *a: {a[i]: a[i + 1]
                                 for i in range(0, len(a), 2)},
    "hasKey": lambda r, d, v, m, k: isinstance(m, dict) and k in m,
    "include": lambda r, d, v, name, ctx: r.include(name, ctx),
    "tpl": lambda r, d, v, s, ctx: r.render(s),
    "required": lambda r, d, v, msg, x: x,
    "trunc": lambda r, d, v, n, x: _render_scalar(x)[:int(n)],
    "trimSuffix": lambda r, d, v, suf, x: _render_scalar(x).rstrip(suf),
    "index": lambda r, d, v, c, *keys: _index(c, keys),
    "int": lambda r, d, v, x: int(float(x)) if x not in (UNDEF, None, "")
    else 0,
    "float": lambda r, d, v, x: float(x) if x not in (UNDEF, None, "")
    else 0.0,
    "toString": lambda r, d, v, x: _render_scalar(x),
}


def _index(c, keys):
    cur = c
    for k in keys:
        if isinstance(cur, dict):
            cur = cur.get(k, UNDEF)
        elif isinstance(cur, list):
            cur = cur[int(k)] if int(k) < len(cur) else UNDEF
        else:
            return UNDEF
    return cur


def _go_printf(fmt: str, args) -> str:
    return re.sub(r"%[sdv]", lambda m, it=iter(args):
                  _render_scalar(next(it)), fmt)


# ---- chart rendering -------------------------------------------------------
def _deep_merge(base: Dict, over: Dict) -> Dict:
    out = dict(base)
    for k, v in over.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = v
    return out


def load_values(overrides: Optional[Dict] = None) -> Dict:
    with open(os.path.join(CHART_DIR, "values.yaml")) as f:
        vals = yaml.safe_load(f) or {}
    return _deep_merge(vals, overrides or {})


def _set_path(values: Dict, path: str, val: Any) -> None:
    """helm-unittest `set:` path syntax: a.b[0].c"""
    toks = re.findall(r"[^.\[\]]+|\[\d+\]", path)
    cur: Any = values
    for i, t in enumerate(toks):
        last = i == len(toks) - 1
        if t.startswith("["):
            idx = int(t[1:-1])
            while len(cur) <= idx:
                cur.append({})
            if last:
                cur[idx] = val
            else:
                cur = cur[idx]
        else:
            if last:
                cur[t] = val
            else:
                nxt_is_idx = i + 1 < len(toks) and toks[i + 1].startswith("[")
                if t not in cur or cur[t] is None:
                    cur[t] = [] if nxt_is_idx else {}
                cur = cur[t]


def render_template(template: str, values: Optional[Dict] = None,
                    release: str = "release",
                    namespace: str = "default") -> List[Dict]:
    """Render one chart template file -> list of YAML documents."""
    vals = values if values is not None else load_values()
    r = Renderer(vals, release, namespace)
    # load helpers (defines)
    helpers = os.path.join(CHART_DIR, "templates", "_helpers.tpl")
    if os.path.exists(helpers):
        with open(helpers) as f:
            r.exec_nodes(r.parse(f.read()), r.root, {}, [])
    with open(os.path.join(CHART_DIR, "templates", template)) as f:
        text = r.render(f.read())
    docs = [d for d in yaml.safe_load_all(text) if d]
    return docs


# ---- helm-unittest spec runner --------------------------------------------
def _get_path(doc: Any, path: str) -> Any:
    toks = re.findall(r'"[^"]+"|[^.\[\]]+|\[\d+\]', path)
    cur = doc
    for t in toks:
        if cur is None:
            return None
        if t.startswith("["):
            idx = int(t[1:-1])
            cur = cur[idx] if isinstance(cur, list) and idx < len(cur) \
                else None
        else:
            key = t.strip('"')
            cur = cur.get(key) if isinstance(cur, dict) else None
    return cur


def run_unittest_spec(spec_path: str) -> List[str]:
    """Execute one helm-unittest YAML spec; returns a list of failure
    strings (empty = all assertions passed)."""
    with open(spec_path) as f:
        spec = yaml.safe_load(f)
    failures: List[str] = []
    templates = spec.get("templates", [])
    for test in spec.get("tests", []):
        overrides: Dict = {}
        for path, val in (test.get("set") or {}).items():
            _set_path(overrides, path, val)
        values = load_values()
        for path, val in (test.get("set") or {}).items():
            _set_path(values, path, val)
        docs: List[Dict] = []
        err: Optional[str] = None
        try:
            for t in templates:
                docs.extend(render_template(t, values))
        except Exception as e:  # template error
            err = str(e)
        for a in test.get("asserts", []):
            di = a.get("documentIndex", 0)
            doc = docs[di] if di < len(docs) else None
            name = test.get("it", "?")

            def fail(msg: str) -> None:
                failures.append(f"{os.path.basename(spec_path)}: "
                                f"{name}: {msg}")

            if err:
                fail(f"render error: {err}")
                break
            if "hasDocuments" in a:
                if len(docs) != a["hasDocuments"]["count"]:
                    fail(f"hasDocuments {a['hasDocuments']['count']} != "
                         f"{len(docs)}")
                continue
            if doc is None:
                fail(f"no document at index {di} (have {len(docs)})")
                continue
            if "isKind" in a:
                if doc.get("kind") != a["isKind"]["of"]:
                    fail(f"kind {doc.get('kind')} != {a['isKind']['of']}")
            elif "equal" in a:
                got = _get_path(doc, a["equal"]["path"])
                if got != a["equal"]["value"]:
                    fail(f"equal {a['equal']['path']}: {got!r} != "
                         f"{a['equal']['value']!r}")
            elif "notEqual" in a:
                got = _get_path(doc, a["notEqual"]["path"])
                if got == a["notEqual"]["value"]:
                    fail(f"notEqual {a['notEqual']['path']}: both "
                         f"{got!r}")
            elif "contains" in a:
                got = _get_path(doc, a["contains"]["path"])
                want = a["contains"]["content"]
                if not isinstance(got, list) or want not in got:
                    fail(f"contains {a['contains']['path']}: {want!r} "
                         f"not in {got!r}")
            elif "notContains" in a:
                got = _get_path(doc, a["notContains"]["path"])
                want = a["notContains"]["content"]
                if isinstance(got, list) and want in got:
                    fail(f"notContains {a['notContains']['path']}: "
                         f"{want!r} present")
            elif "exists" in a:
                if _get_path(doc, a["exists"]["path"]) is None:
                    fail(f"exists {a['exists']['path']}: missing")
            elif "notExists" in a:
                if _get_path(doc, a["notExists"]["path"]) is not None:
                    fail(f"notExists {a['notExists']['path']}: present")
            elif "isNull" in a:
                if _get_path(doc, a["isNull"]["path"]) is not None:
                    fail(f"isNull {a['isNull']['path']}: not null")
            elif "isNotEmpty" in a:
                got = _get_path(doc, a["isNotEmpty"]["path"])
                if not got:
                    fail(f"isNotEmpty {a['isNotEmpty']['path']}: empty")
            elif "matchRegex" in a:
                got = _get_path(doc, a["matchRegex"]["path"])
                if not re.search(a["matchRegex"]["pattern"],
                                 str(got or "")):
                    fail(f"matchRegex {a['matchRegex']['path']}")
            else:
                # an unknown assertion type must never silently pass
                fail(f"unsupported assertion {sorted(a.keys())}")
    return failures
