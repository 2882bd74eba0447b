"""Flag-surface tests: the reference's tf.app.flags call shape parses."""

from rocps import config


def _fresh():
    return config._FlagValues()


def test_reference_flag_surface():
    f = _fresh()
    f._define("ps_hosts", "", "", str)
    f._define("worker_hosts", "", "", str)
    f._define("job_name", "", "", str)
    f._define("task_index", 0, "", int)
    f._define("issync", False, "", bool)
    f._define("learning_rate", 0.01, "", float)
    rest = f._parse([
        "--ps_hosts=h1:2222,h2:2222", "--worker_hosts", "h3:2223",
        "--job_name=worker", "--task_index", "1", "--issync=1",
        "--learning_rate=0.5", "positional_left"])
    assert f.ps_hosts == "h1:2222,h2:2222"
    assert f.worker_hosts == "h3:2223"
    assert f.job_name == "worker"
    assert f.task_index == 1
    assert f.issync is True
    assert f.learning_rate == 0.5
    assert rest == ["positional_left"]


def test_bool_forms():
    f = _fresh()
    f._define("issync", False, "", bool)
    for v, expect in [("1", True), ("true", True), ("True", True),
                      ("0", False), ("false", False)]:
        f._parse([f"--issync={v}"])
        assert f.issync is expect, v
    f._parse(["--issync"])  # bare flag
    assert f.issync is True


def test_defaults_without_parse():
    f = _fresh()
    f._define("train_steps", 100, "", int)
    f._parse([])
    assert f.train_steps == 100
