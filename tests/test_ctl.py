"""amd-dra-ctl CLI tests on the fake HAL."""

import json

from k8s_dra_driver_amd.ctl import main


def test_list(capsys):
    assert main(["--hal", "fake", "list"]) == 0
    out = capsys.readouterr().out
    assert "gpu-0: AMD Instinct MI355X [gfx950]" in out
    assert "mode=SPX/NPS1" in out


def test_topology_matrix(capsys):
    assert main(["--hal", "fake", "topology"]) == 0
    out = capsys.readouterr().out
    assert "hives:" in out
    assert out.count("x") >= 8  # diagonal


def test_slice_json(capsys):
    assert main(["--hal", "fake", "slice"]) == 0
    out = json.loads(capsys.readouterr().out)
    assert len(out["devices"]) == 8
    assert (
        out["devices"][0]["basic"]["attributes"]["gpu.amd.com/type"]["string"]
        == "gpu"
    )


def test_partition_and_list(capsys):
    assert main(["--hal", "fake", "partition", "0", "cpx"]) == 0
    out = capsys.readouterr().out
    assert "switched to CPX/NPS1 (8 device(s))" in out


def test_partition_invalid(capsys):
    assert main(["--hal", "fake", "partition", "0", "SPX", "NPS4"]) == 1
    assert "refused" in capsys.readouterr().err


def test_health(capsys):
    assert main(["--hal", "fake", "health"]) == 0
    assert "healthy" in capsys.readouterr().out


def test_ctl_profile_fetches_diag(tmp_path):
    from k8s_dra_driver_amd.ctl import main
    from k8s_dra_driver_amd.utils.diag import DiagServer

    srv = DiagServer(0, host="127.0.0.1")
    srv.start()
    try:
        import contextlib
        import io

        buf = io.StringIO()
        with contextlib.redirect_stdout(buf):
            rc = main(
                ["profile", "--port", str(srv.port), "--seconds", "0.2"]
            )
        assert rc == 0
        assert buf.getvalue().startswith("# cpu profile")
    finally:
        srv.stop()


def test_slice_prospective(capsys):
    assert main(["--hal", "fake", "slice", "--prospective", "cpx"]) == 0
    out = json.loads(capsys.readouterr().out)
    assert len(out["devices"]) == 8 + 64
    assert len(out["sharedCounters"]) == 8


def test_labels_preview(capsys):
    assert main(["--hal", "fake", "labels"]) == 0
    out = capsys.readouterr().out
    assert "gpu.amd.com/gpu.count=8" in out
    assert "gpu.amd.com/gpu.architecture=gfx950" in out
