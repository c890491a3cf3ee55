from multiverso_amd.configure import (define_flag, get_flag, parse_cmd_flags,
                                      set_flag)


def test_defaults():
    assert get_flag("updater_type") == "default"
    assert get_flag("sync") is False
    assert get_flag("omp_threads") == 0  # 0 = torch default thread pool


def test_parse_cmd_flags_consumes_known():
    set_flag("sync", False)
    rest = parse_cmd_flags(["prog", "-sync=true", "-updater_type=sgd",
                            "-notaflag=1", "positional"])
    assert rest == ["prog", "-notaflag=1", "positional"]
    assert get_flag("sync") is True
    assert get_flag("updater_type") == "sgd"
    set_flag("sync", False)
    set_flag("updater_type", "default")


def test_set_flag_typed():
    set_flag("omp_threads", "8")
    assert get_flag("omp_threads") == 8
    set_flag("omp_threads", 4)
    define_flag("my_new_flag", 1.5)
    set_flag("my_new_flag", "2.5")
    assert get_flag("my_new_flag") == 2.5
