from trtlab_amd.utils import bytes_to_string, round_up, string_to_bytes


def test_string_to_bytes():
    assert string_to_bytes("10MiB") == 10 * 1024 * 1024
    assert string_to_bytes("1kb") == 1000
    assert string_to_bytes("512") == 512
    assert string_to_bytes("1.5GiB") == int(1.5 * 1024**3)


def test_bytes_to_string():
    assert bytes_to_string(512) == "512 B"
    assert bytes_to_string(10 * 1024 * 1024) == "10.0 MiB"


def test_round_up():
    assert round_up(1, 256) == 256
    assert round_up(256, 256) == 256
    assert round_up(257, 256) == 512
