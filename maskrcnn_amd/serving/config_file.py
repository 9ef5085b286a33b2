"""TF-Serving server config files.

TF-Serving's ModelServer takes `--model_config_file=models.config` (a
`ModelServerConfig` text proto listing served models) and
`--enable_batching --batching_parameters_file=batching_parameters.txt`
(a `SessionBundleConfig.BatchingParameters` text proto). Both are
protobuf TEXT format; this module parses the subset those two files
use — `key: value` scalars, `key { ... }` messages, repeated fields —
with no protobuf dependency.

models.config:
    model_config_list {
      config { name: 'mask_rcnn'  base_path: '/models/mask_rcnn'
               model_platform: 'tensorflow' }
      config { name: 'other'      base_path: '/models/other' }
    }

batching_parameters.txt (wrapped-scalar fields):
    max_batch_size { value: 32 }
    batch_timeout_micros { value: 2000 }
    num_batch_threads { value: 8 }
    max_enqueued_batches { value: 1000000 }
"""

from __future__ import annotations

import re
from typing import Dict, List

_TOKEN = re.compile(
    r"""[{}]|:|[A-Za-z_]\w*|'(?:[^'\\]|\\.)*'|"(?:[^"\\]|\\.)*"|"""
    r"""-?\d+(?:\.\d+)?(?:[eE][-+]?\d+)?""")


def _tokenize(text: str) -> List[str]:
    # Strip '#' comments (text-proto comment syntax) line by line.
    lines = [ln.split("#", 1)[0] for ln in text.splitlines()]
    return _TOKEN.findall("\n".join(lines))


def _value(tok: str):
    if tok[:1] in "'\"":
        return tok[1:-1].replace("\\'", "'").replace('\\"', '"')
    try:
        return int(tok)
    except ValueError:
        pass
    try:
        return float(tok)
    except ValueError:
        pass
    if tok in ("true", "false"):
        return tok == "true"
    return tok


def parse_text_proto(text: str) -> Dict:
    """Parse text-proto into nested dicts; repeated fields -> lists."""
    tokens = _tokenize(text)
    pos = 0

    def parse_message(depth=0):
        nonlocal pos
        msg: Dict = {}

        def put(key, val):
            if key in msg:
                if not isinstance(msg[key], list):
                    msg[key] = [msg[key]]
                msg[key].append(val)
            else:
                msg[key] = val

        while pos < len(tokens):
            tok = tokens[pos]
            if tok == "}":
                pos += 1
                return msg
            key = tok
            pos += 1
            if pos < len(tokens) and tokens[pos] == ":":
                pos += 1
            if pos < len(tokens) and tokens[pos] == "{":
                pos += 1
                put(key, parse_message(depth + 1))
            else:
                if pos >= len(tokens):
                    raise ValueError(f"dangling field {key!r}")
                put(key, _value(tokens[pos]))
                pos += 1
        if depth:
            raise ValueError("unbalanced braces in text proto")
        return msg

    return parse_message()


def parse_model_config(path: str) -> List[Dict]:
    """models.config -> [{'name', 'base_path', 'model_platform'}, ...]."""
    d = parse_text_proto(open(path).read())
    cfgs = d.get("model_config_list", {}).get("config", [])
    if isinstance(cfgs, dict):
        cfgs = [cfgs]
    out = []
    for c in cfgs:
        if "name" not in c or "base_path" not in c:
            raise ValueError(f"model config missing name/base_path: {c}")
        entry = {"name": str(c["name"]),
                 "base_path": str(c["base_path"]),
                 "model_platform": str(c.get("model_platform",
                                             "tensorflow"))}
        # model_version_policy { all {} } / { latest { num_versions: 1 } }
        mvp = c.get("model_version_policy")
        if isinstance(mvp, dict) and "all" in mvp:
            entry["version_policy"] = "all"
        # version_labels { key: 'stable' value: 2 } (repeated) — the
        # TF-Serving canary/rollback mechanism: clients address
        # ModelSpec.version_label instead of a number.
        vl = c.get("version_labels")
        if vl:
            if isinstance(vl, dict):
                vl = [vl]
            entry["version_labels"] = {
                str(e["key"]): int(e["value"]) for e in vl
                if "key" in e and "value" in e}
        out.append(entry)
    return out


def parse_batching_parameters(path: str) -> Dict[str, int]:
    """batching_parameters.txt -> plain ints for the known fields."""
    d = parse_text_proto(open(path).read())
    out = {}
    for k in ("max_batch_size", "batch_timeout_micros",
              "num_batch_threads", "max_enqueued_batches"):
        if k in d:
            v = d[k]
            out[k] = int(v["value"] if isinstance(v, dict) else v)
    return out
