#!/usr/bin/env python3
"""Sweep driver — the equivalent of the reference's ``batch_run.py``: a
FileLock-guarded batch JSON holds per-config run counts and nested overrides;
each invocation decrements a count, merges the override into the base config
and dispatches to obj.main or nsra.main by run-name substring
(reference ``batch_run.py:31-62``).

  python examples/batch_run.py configs/batch.json
"""
import json
import os
import sys

from filelock import FileLock

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.config import AttrDict, load_config, merge_override, parse_args

import nsra  # noqa: E402  (sibling entry modules)
import obj  # noqa: E402


def claim_run(batch_file: str):
    """Atomically pick a run with remaining count and decrement it."""
    with FileLock(batch_file + ".lock"):
        with open(batch_file) as f:
            batch = json.load(f)
        for name, spec in batch["runs"].items():
            if spec.get("count", 0) > 0:
                spec["count"] -= 1
                with open(batch_file, "w") as g:
                    json.dump(batch, g, indent=2)
                return name, batch["base_config"], spec.get("overrides", {})
    return None, None, None


def main(batch_file: str):
    while True:
        name, base_cfg_file, overrides = claim_run(batch_file)
        if name is None:
            print("batch complete")
            return
        base = load_config(os.path.join(os.path.dirname(batch_file), base_cfg_file)
                           if not os.path.isabs(base_cfg_file) else base_cfg_file)
        cfg = AttrDict(merge_override(base.to_dict(), overrides))
        cfg.general.name = name
        print(f"dispatching run {name!r}")
        if "nsr" in name or "ns-" in name:
            nsra.main(cfg)
        else:
            obj.main(cfg)


if __name__ == "__main__":
    main(parse_args())
