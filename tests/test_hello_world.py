"""Hello-world plumbing smoke (BASELINE config #1 shape)."""

from cosmos_curate_amd.pipelines.examples.hello_world_pipeline import (
    EXAMPLE_PROMPTS,
    main,
)


def test_hello_world_pipeline(capsys):
    out = main()
    assert len(out) == len(EXAMPLE_PROMPTS)
    for task, prompt in zip(out, EXAMPLE_PROMPTS):
        assert task.prompt == prompt.lower()
        assert task.output is not None and task.output.startswith(prompt.lower())
    printed = capsys.readouterr().out
    for prompt in EXAMPLE_PROMPTS:
        assert prompt.lower() in printed
