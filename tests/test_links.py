from bee2bee_amd.mesh.links import generate_join_link, parse_join_link


def test_roundtrip():
    link = generate_join_link(
        "connectit", "llama3-8b", "ab" * 32, ["ws://10.0.0.1:4001", "wss://x.io:443"]
    )
    parsed = parse_join_link(link)
    assert parsed["network"] == "connectit"
    assert parsed["model"] == "llama3-8b"
    assert parsed["hash"] == "ab" * 32
    assert parsed["bootstrap"] == ["ws://10.0.0.1:4001", "wss://x.io:443"]


def test_no_bootstrap():
    link = generate_join_link("n", "m", "h", [])
    assert parse_join_link(link)["bootstrap"] == []


def test_reference_scheme_accepted():
    # links from the reference implementation use the same scheme set
    link = "coithub.org://join?network=n&model=m&hash=h"
    assert parse_join_link(link)["model"] == "m"


def test_invalid():
    import pytest

    with pytest.raises(ValueError):
        parse_join_link("https://example.com/join?x=1")
