"""_signal unit tests: exit flag + the in-place rescale directive
channel (the e2e behavior is covered by the controller drills; these
pin the low-level semantics)."""

import json
import os

import adaptdl_amd._signal as _signal


def test_exit_flag_set_and_programmatic():
    _signal.set_exit_flag(False)
    assert not _signal.get_exit_flag()
    _signal.set_exit_flag(True)
    assert _signal.get_exit_flag()
    _signal.set_exit_flag(False)


def test_rescale_request_lifecycle(tmp_path, monkeypatch):
    from adaptdl_amd.torch import _rejoin

    monkeypatch.setenv("ADAPTDL_CHECKPOINT_PATH", str(tmp_path))
    monkeypatch.setattr(_signal, "_RESCALE_SEEN", False)
    monkeypatch.setattr(_rejoin, "_applied_version", 0)

    # No signal yet: nothing pending even if a file exists.
    with open(tmp_path / "rescale-inplace.json", "w") as f:
        json.dump({"version": 1, "world": 1, "master_port": 1234}, f)
    assert _signal.get_rescale_request() == (0, None)

    # Signal received: the directive becomes visible.
    _signal._usr2_handler(None, None)
    ver, directive = _signal.get_rescale_request()
    assert ver == 1
    assert directive["world"] == 1

    # Already-applied versions read as not-pending.
    monkeypatch.setattr(_rejoin, "_applied_version", 1)
    assert _signal.get_rescale_request() == (0, None)

    # A newer directive supersedes.
    with open(tmp_path / "rescale-inplace.json", "w") as f:
        json.dump({"version": 2, "world": 1, "master_port": 4321}, f)
    ver, directive = _signal.get_rescale_request()
    assert ver == 2 and directive["master_port"] == 4321

    # Corrupt file: treated as not-pending, no exception.
    (tmp_path / "rescale-inplace.json").write_text("{nope")
    monkeypatch.setattr(_rejoin, "_applied_version", 0)
    assert _signal.get_rescale_request() == (0, None)


def test_handler_marker_written(tmp_path, monkeypatch):
    """install_signal_handlers writes the SIGUSR2 readiness marker the
    controller gates its signals on."""
    import signal as signal_mod
    saved = {s: signal_mod.getsignal(s)
             for s in (signal_mod.SIGTERM, signal_mod.SIGINT,
                       signal_mod.SIGUSR2)}
    monkeypatch.setenv("ADAPTDL_CHECKPOINT_PATH", str(tmp_path))
    monkeypatch.setenv("ADAPTDL_REPLICA_RANK", "3")
    monkeypatch.setattr(_signal, "_INSTALLED", False)
    try:
        _signal.install_signal_handlers()
        assert (tmp_path / ".sigusr2-ready-3").exists()
    finally:
        for s, h in saved.items():
            signal_mod.signal(s, h)
        _signal._INSTALLED = False
