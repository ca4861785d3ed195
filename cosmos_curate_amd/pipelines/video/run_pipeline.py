"""Unified pipeline entry (reference run_pipeline.py CLI-mode subset).

    python -m cosmos_curate_amd.pipelines.video.run_pipeline split --input-video-path ...
    python -m cosmos_curate_amd.pipelines.video.run_pipeline dedup --input-embeddings-path ...
    python -m cosmos_curate_amd.pipelines.video.run_pipeline shard --input-clip-path ...

Also accepts a .json/.yaml config file as the first argument (the
reference's config mode, run_pipeline.py:17-27): the file holds
``{"pipeline": "split", "args": {"input_video_path": ...}}``; keys use
underscores and map 1:1 onto the subcommand flags.
"""

from __future__ import annotations

import json
import pathlib
import sys

_CONFIG_EXTENSIONS = frozenset({".json", ".yaml", ".yml"})

_ENTRY_POINTS = {
    "split": ("cosmos_curate_amd.pipelines.video.splitting_pipeline", "cli_run_split"),
    "dedup": ("cosmos_curate_amd.pipelines.video.dedup_pipeline", "cli_run_dedup"),
    "shard": ("cosmos_curate_amd.pipelines.video.sharding_pipeline", "cli_run_shard"),
}


def _config_to_argv(path: pathlib.Path) -> tuple[str, list[str]]:
    text = path.read_text()
    if path.suffix in {".yaml", ".yml"}:
        import yaml

        cfg = yaml.safe_load(text)
    else:
        cfg = json.loads(text)
    pipeline = cfg["pipeline"]
    argv: list[str] = []
    for key, value in cfg.get("args", {}).items():
        flag = "--" + key.replace("_", "-")
        if isinstance(value, bool):
            if value:
                argv.append(flag)
        else:
            argv.extend([flag, str(value)])
    return pipeline, argv


def main(argv: list[str] | None = None):
    argv = list(sys.argv[1:] if argv is None else argv)
    if not argv or argv[0] in {"-h", "--help"}:
        names = "|".join(_ENTRY_POINTS)
        print(f"usage: run_pipeline <{names}> [flags] | run_pipeline config.(json|yaml)",
              file=sys.stderr)
        raise SystemExit(0 if argv else 2)
    first = pathlib.Path(argv[0])
    if first.suffix in _CONFIG_EXTENSIONS:
        pipeline, rest = _config_to_argv(first)
        rest += argv[1:]
    else:
        pipeline, rest = argv[0], argv[1:]
    if pipeline not in _ENTRY_POINTS:
        msg = f"unknown pipeline {pipeline!r} (have: {sorted(_ENTRY_POINTS)})"
        raise SystemExit(msg)
    import importlib

    mod_name, fn_name = _ENTRY_POINTS[pipeline]
    fn = getattr(importlib.import_module(mod_name), fn_name)
    return fn(rest)


if __name__ == "__main__":
    main()
