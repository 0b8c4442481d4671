"""parse_uri vector parity with the reference's gtest corpus
(reference src/main/cpp/tests/parse_uri.cpp — SIMPLE / SPARK_EDGES / IPv6 /
IPv4 / UTF8 / QUERY suites; two rows with source-encoding-ambiguous raw
bytes are omitted). Each tuple: (uri, protocol, host, query, path) with
None = null result."""
import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType

# (uri, protocol, host, query, path)
SIMPLE = [
    ("https://www.nvidia.com/s/uri?param1=2", "https", "www.nvidia.com",
     "param1=2", "/s/uri"),
    ("http://www.nvidia.com", "http", "www.nvidia.com", None, ""),
    ("file://path/to/a/cool/file", "file", "path", None, "/to/a/cool/file"),
    ("smb://network/path/to/file", "smb", "network", None, "/path/to/file"),
    ("http:/www.nvidia.com", "http", None, None, "/www.nvidia.com"),
    ("file:path/to/a/cool/file", "file", None, None, None),
    ("/network/path/to/file", None, None, None, "/network/path/to/file"),
    ("nvidia.com", None, None, None, "nvidia.com"),
    ("www.nvidia.com/s/uri", None, None, None, "www.nvidia.com/s/uri"),
]

SPARK_EDGES = [
    ("https://nvidia.com/https&#://nvidia.com", "https", "nvidia.com",
     None, "/https&"),
    ("https://http://www.nvidia.com", "https", "http", None,
     "//www.nvidia.com"),
    ("filesystemmagicthing://bob.yaml", "filesystemmagicthing", "bob.yaml",
     None, ""),
    ("nvidia.com:8080", "nvidia.com", None, None, None),
    ("http://thisisinvalid.data/due/to-the_character%s/inside*the#url`~",
     None, None, None, None),
    ("file:/absolute/path", "file", None, None, "/absolute/path"),
    ("//www.nvidia.com", None, "www.nvidia.com", None, ""),
    ("#bob", None, None, None, ""),
    ("#this%doesnt#make//sense://to/me", None, None, None, None),
    ("HTTP:&bob", "HTTP", None, None, None),
    ("/absolute/path", None, None, None, "/absolute/path"),
    ("http://%77%77%77.%4EV%49%44%49%41.com", "http", None, None, ""),
    ("https:://broken.url", "https", None, None, None),
    ("https://www.nvidia.com/q/This%20is%20a%20query", "https",
     "www.nvidia.com", None, "/q/This%20is%20a%20query"),
    (b"https://www.nvidia.com/\x93path/path/to/file", None, None, None,
     None),
    ("http://?", "http", None, "", ""),
    ("http://??", "http", None, "?", ""),
    ("http://??/", "http", None, "?/", ""),
    ("http://#", "http", None, None, ""),
    ("http://user:pass@host/file;param?query;p2", "http", "host",
     "query;p2", "/file;param"),
    ("http://[1:2:3:4:5:6:7::]", "http", "[1:2:3:4:5:6:7::]", None, ""),
    ("http://[::2:3:4:5:6:7:8]", "http", "[::2:3:4:5:6:7:8]", None, ""),
    ("http://[fe80::7:8%eth0]", "http", "[fe80::7:8%eth0]", None, ""),
    ("http://[fe80::7:8%1]", "http", "[fe80::7:8%1]", None, ""),
    ("http://foo.bar/abc/" + "\\" * 3 + "http://foo.bar/abc.gif" +
     "\\" * 3, None, None, None, None),
    ("www.nvidia.com:8100/servlet/impc.DisplayCredits?primekey_in="
     "2000041100:05:14115240636", "www.nvidia.com", None, None, None),
    ("https://nvidia.com/2Ru15Ss ", None, None, None, None),
    ("http://www.nvidia.com/plugins//##", None, None, None, None),
    ("www.nvidia.com:81/Free.fr/L7D9qw9X4S-aC0&amp;D4X0/Panels&amp;"
     "solutionId=0X54a/cCdyncharset=UTF-8&amp;t=01wx58Tab&amp;ps=solution/"
     "ccmd=_help&amp;locale0X1&amp;countrycode=MA/", "www.nvidia.com",
     None, None, None),
    ("http://www.nvidia.com//wp-admin/includes/index.html#9389#123",
     None, None, None, None),
    ("http://-.~_!$&'()*+,;=:%40:80%2f::::::@nvidia.com:443", "http",
     "nvidia.com", None, ""),
    ("http://userid:password@example.com:8080/", "http", "example.com",
     None, "/"),
    ("http://.www.nvidia.com./", "http", None, None, "/"),
    ("http://www.nvidia..com/", "http", None, None, "/"),
]

IPV6 = [
    ("https://[fe80::]", "https", "[fe80::]", None, ""),
    ("https://[2001:0db8:85a3:0000:0000:8a2e:0370:7334]", "https",
     "[2001:0db8:85a3:0000:0000:8a2e:0370:7334]", None, ""),
    ("https://[2001:0DB8:85A3:0000:0000:8A2E:0370:7334]", "https",
     "[2001:0DB8:85A3:0000:0000:8A2E:0370:7334]", None, ""),
    ("https://[2001:db8::1:0]", "https", "[2001:db8::1:0]", None, ""),
    ("http://[2001:db8::2:1]", "http", "[2001:db8::2:1]", None, ""),
    ("https://[::1]", "https", "[::1]", None, ""),
    ("https://[2001:db8:85a3:8d3:1319:8a2e:370:7348]:443", "https",
     "[2001:db8:85a3:8d3:1319:8a2e:370:7348]", None, ""),
    ("https://[2001:db8:3333:4444:5555:6666:1.2.3.4]/path/to/file",
     "https", "[2001:db8:3333:4444:5555:6666:1.2.3.4]", None,
     "/path/to/file"),
    ("https://[2001:db8:3333:4444:5555:6666:7777:8888:1.2.3.4]"
     "/path/to/file", None, None, None, None),
    ("https://[::db8:3333:4444:5555:6666:1.2.3.4]/path/to/file]",
     None, None, None, None),
]

IPV4 = [
    ("https://192.168.1.100/", "https", "192.168.1.100", None, "/"),
    ("https://192.168.1.100:8443/", "https", "192.168.1.100", None, "/"),
    ("https://192.168.1.100.5/", "https", None, None, "/"),
    ("https://192.168.1/", "https", None, None, "/"),
    ("https://280.100.1.1/", "https", None, None, "/"),
    ("https://182.168..100/path/to/file", "https", None, None,
     "/path/to/file"),
]

UTF8 = [
    ("https://nvidia.com/%4EV%49%44%49%41", "https", "nvidia.com", None,
     "/%4EV%49%44%49%41"),
    ("http://%77%77%77.%4EV%49%44%49%41.com", "http", None, None, ""),
    ("http://✪↩d⁚f„⁈.ws/123", "http", None, None, "/123"),
    ("https:// /path/to/file", None, None, None, None),
]

QUERY_URIS = [
    "https://www.nvidia.com/path?param0=1&param2=3&param4=5",
    "https:// /?params=5&cloth=0&metal=1&param0=param3",
    "https://[2001:db8::2:1]:443/parms/in/the/uri?a=b&param0=true",
    "https://[::1]/?invalid=param&f„⁈.=7&param0=3",
    "https://[::1]/?invalid=param&param0=f„⁈&~.=!@&^",
    "userinfo@www.nvidia.com/path?query=1&param0=5#Ref",
    "https://www.nvidia.com/path?brokenparam0=1&fakeparam0=5&param0=true",
    "http://nvidia.com?CBA=CBA&C=C",
]
QUERY_FULL = ["param0=1&param2=3&param4=5", None, "a=b&param0=true",
              "invalid=param&f„⁈.=7&param0=3", None, "query=1&param0=5",
              "brokenparam0=1&fakeparam0=5&param0=true", "CBA=CBA&C=C"]
QUERY_PARAM0 = ["1", None, "true", "3", None, "5", "true", None]
QUERY_C = [None, None, None, None, None, None, None, "C"]
QUERY_COLS = ["param0", "q", "a", "invalid", "test", "query",
              "fakeparam0", "C"]
QUERY_COL_EXP = ["1", None, "b", "param", None, "1", "5", "C"]


def _col(uris):
    return Column.from_pylist(list(uris), DType.STRING, "cuda")


def _run(uris, part, key=""):
    from spark_rapids_jni_amd.ops.sketch import UriPart, parse_uri
    return parse_uri(_col(uris), part, key).to_pylist()


@pytest.mark.gpu
@pytest.mark.parametrize("suite", ["SIMPLE", "SPARK_EDGES", "IPV6", "IPV4",
                                   "UTF8"])
def test_parse_uri_suites(suite):
    from spark_rapids_jni_amd.ops.sketch import UriPart
    data = globals()[suite]
    uris = [r[0] for r in data]
    for part, idx in ((UriPart.PROTOCOL, 1), (UriPart.HOST, 2),
                      (UriPart.QUERY, 3), (UriPart.PATH, 4)):
        got = _run(uris, part)
        exp = [r[idx] for r in data]
        for i, (g, x) in enumerate(zip(got, exp)):
            assert g == x, (f"{suite} {part.name} row {i} "
                            f"{uris[i]!r}: {g!r} != {x!r}")


@pytest.mark.gpu
def test_parse_uri_query_keys():
    from spark_rapids_jni_amd.ops.sketch import UriPart
    assert _run(QUERY_URIS, UriPart.QUERY) == QUERY_FULL
    assert _run(QUERY_URIS, UriPart.QUERY_KEY, "param0") == QUERY_PARAM0
    assert _run(QUERY_URIS, UriPart.QUERY_KEY, "C") == QUERY_C
    key_col = _col(QUERY_COLS)
    got = _run(QUERY_URIS, UriPart.QUERY_KEY, key_col)
    assert got == QUERY_COL_EXP


@pytest.mark.gpu
def test_parse_uri_null_input():
    from spark_rapids_jni_amd.ops.sketch import UriPart
    got = _run(["https://a.com", None], UriPart.PROTOCOL)
    assert got == ["https", None]
