"""Peer list / host list / cluster plan tests (reference:
srcs/go/plan/{cluster,hostspec}_test.go)."""
import json

from kungfu_amd import _core


def test_gen_peer_list():
    pl = _core.gen_peer_list("127.0.0.1:4", 4, 31100)
    assert pl == "127.0.0.1:31100,127.0.0.1:31101,127.0.0.1:31102," \
                 "127.0.0.1:31103"


def test_gen_peer_list_multi_host():
    pl = _core.gen_peer_list("10.0.0.1:2,10.0.0.2:2", 3, 31100)
    assert pl == "10.0.0.1:31100,10.0.0.1:31101,10.0.0.2:31100"


def test_gen_runner_list():
    rl = _core.gen_runner_list("10.0.0.1:2,10.0.0.2:2", 38080)
    assert rl == "10.0.0.1:38080,10.0.0.2:38080"


def test_cluster_resize_grow_least_loaded():
    c = {
        "runners": ["10.0.0.1:38080", "10.0.0.2:38080"],
        "workers": ["10.0.0.1:31100", "10.0.0.1:31101", "10.0.0.2:31100"],
    }
    out = json.loads(_core.cluster_resize_json(json.dumps(c), 4, 31100))
    assert out["workers"][:3] == c["workers"]
    # new worker goes to the least-loaded host (10.0.0.2) on a fresh port
    assert out["workers"][3] == "10.0.0.2:31101"


def test_cluster_resize_shrink_keeps_prefix():
    c = {
        "runners": ["10.0.0.1:38080"],
        "workers": ["10.0.0.1:31100", "10.0.0.1:31101", "10.0.0.1:31102"],
    }
    out = json.loads(_core.cluster_resize_json(json.dumps(c), 2, 31100))
    assert out["workers"] == c["workers"][:2]
