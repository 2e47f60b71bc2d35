"""Log plumbing (tail-while-running, regex filter, streams) and events
handler pluggability (parity: local_scheduler LogIterator tests,
runner/events handler registry)."""

import threading
import time

import pytest

from torchx_amd.runner import get_runner
from torchx_amd.specs import AppState


def _run(runner, args, tmp_path, **kw):
    return runner.run_component(
        "utils.sh", args, scheduler="local_cwd",
        cfg={"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False},
        **kw,
    )


def test_log_tail_while_running(tmp_path):
    with get_runner("t") as runner:
        handle = _run(
            runner,
            ["bash", "-c",
             "for i in 1 2 3 4 5; do echo line-$i; sleep 0.3; done"],
            tmp_path,
        )
        got = []

        def pull():
            for ln in runner.log_lines(handle, "sh", 0, should_tail=True):
                got.append(ln)

        t = threading.Thread(target=pull, daemon=True)
        t.start()
        status = runner.wait(handle, wait_interval=0.3)
        t.join(timeout=30)
        assert status.state == AppState.SUCCEEDED
        assert sum("line-" in ln for ln in got) == 5, got


def test_log_regex_filter(tmp_path):
    with get_runner("t") as runner:
        handle = _run(
            runner,
            ["bash", "-c", "echo keep-1; echo drop-2; echo keep-3"],
            tmp_path,
        )
        runner.wait(handle, wait_interval=0.3)
        lines = list(runner.log_lines(handle, "sh", 0, regex="keep"))
        assert sum("keep" in ln for ln in lines) == 2
        assert not any("drop" in ln for ln in lines)


def test_events_handler_pluggable():
    import logging

    from torchx_amd.runner import events

    records = []

    class Capture(logging.Handler):
        def emit(self, record):
            records.append(record.getMessage())

    # named-handler registry (reference parity: events/handlers.py)
    events.register_handler("capture", Capture())
    logger = events._trace_logger("capture")
    # force our handler onto the cached logger for the test
    logger.addHandler(events.get_logging_handler("capture"))
    with events.log_event("myapi", "local_cwd", "sess"):
        pass
    assert any("myapi" in r for r in records), records


def test_structured_error_surfaces_in_status(tmp_path):
    """A worker that dies writes TORCHELASTIC_ERROR_FILE; the scheduler's
    describe surfaces it in AppStatus."""
    script = tmp_path / "die.py"
    script.write_text(
        "import json, os, sys\n"
        "ef = os.environ.get('TORCHELASTIC_ERROR_FILE')\n"
        "json.dump({'message': 'intentional-kaboom'}, open(ef, 'w'))\n"
        "sys.exit(3)\n"
    )
    with get_runner("t") as runner:
        handle = runner.run_component(
            "utils.python", ["--script", str(script)],
            scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path),
                 "auto_set_hip_visible_devices": False},
        )
        status = runner.wait(handle, wait_interval=0.3)
    assert status.state == AppState.FAILED
    assert "intentional-kaboom" in status.format()


def test_print_log_lines_merges_with_prefixes():
    import io

    from torchx_amd.utils.log_tee import print_log_lines

    def lines(role, replica):
        return [f"{role}-{replica} line{j}" for j in range(3)]

    buf = io.StringIO()
    print_log_lines([("trainer", 0), ("trainer", 1), ("ps", 0)],
                    lines, stream=buf, colored=False)
    out = buf.getvalue().splitlines()
    assert len(out) == 9
    # every line carries its role/replica prefix; streams fully drained
    assert sum(1 for ln in out if ln.startswith("trainer/0 ")) == 3
    assert sum(1 for ln in out if ln.startswith("trainer/1 ")) == 3
    assert sum(1 for ln in out if ln.startswith("ps/0 ")) == 3
    assert "trainer/0 trainer-0 line2" in out


def test_print_log_lines_colored_prefix_stable():
    import io

    from torchx_amd.utils.log_tee import print_log_lines

    buf = io.StringIO()
    print_log_lines([("w", 0)], lambda r, k: ["x"], stream=buf, colored=True)
    line = buf.getvalue()
    assert "\033[" in line and "w/0" in line


def test_print_log_lines_raises_puller_error():
    import io

    import pytest as _pytest

    from torchx_amd.utils.log_tee import print_log_lines

    def boom(role, replica):
        raise RuntimeError("log source gone")

    with _pytest.raises(RuntimeError, match="log source gone"):
        print_log_lines([("w", 0)], boom, stream=io.StringIO(),
                        colored=False)
