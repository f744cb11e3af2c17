"""The ext-proc gRPC picker must be callable by an EPP/Envoy client:
a raw-bytes gRPC stream of ProcessingRequest messages gets back
ProcessingResponse header mutations naming the chosen endpoint via
x-gateway-destination-endpoint."""

import json

import grpc
import pytest

from production_stack_amd.gateway import extproc
from production_stack_amd.gateway.extproc import (
    ExtProcServicer,
    Picker,
    _ld,
    build_headers_response,
    parse_processing_request,
)


def _encode_headers(headers):
    hvs = b"".join(
        _ld(1, _ld(1, k.encode()) + _ld(3, v.encode()))
        for k, v in headers.items()
    )
    return _ld(2, _ld(1, hvs))  # ProcessingRequest.request_headers


def _encode_body(body: bytes):
    return _ld(3, _ld(1, body))  # ProcessingRequest.request_body


def _decode_set_headers(resp: bytes):
    out = {}
    for fno, wt, v in extproc._fields(resp):
        if fno in (2, 3) and wt == 2:  # Headers/BodyResponse
            for f2, _, v2 in extproc._fields(v):
                if f2 == 1:  # CommonResponse
                    for f3, _, v3 in extproc._fields(v2):
                        if f3 == 3:  # HeaderMutation
                            for f4, _, v4 in extproc._fields(v3):
                                if f4 == 1:  # HeaderValueOption
                                    for f5, _, v5 in extproc._fields(v4):
                                        if f5 == 1:  # HeaderValue
                                            k = val = ""
                                            for f6, _, v6 in \
                                                    extproc._fields(v5):
                                                if f6 == 1:
                                                    k = v6.decode()
                                                elif f6 in (2, 3):
                                                    val = v6.decode()
                                            out[k] = val
    return out


def test_wire_codec_roundtrip():
    msg = _encode_headers({":path": "/v1/completions",
                           "x-gateway-candidate-pods": "a,b"})
    parsed = parse_processing_request(msg)
    assert parsed["kind"] == "request_headers"
    assert parsed["headers"]["x-gateway-candidate-pods"] == "a,b"
    body = _encode_body(b'{"prompt": "hi"}')
    parsed = parse_processing_request(body)
    assert parsed["kind"] == "request_body"
    assert parsed["body"] == b'{"prompt": "hi"}'
    resp = build_headers_response({"x-gateway-destination-endpoint": "a"})
    assert _decode_set_headers(resp) == {
        "x-gateway-destination-endpoint": "a"}


@pytest.fixture()
def extproc_server():
    server = extproc.serve(port=19102, algorithm="prefixaware", block=False)
    yield "127.0.0.1:19102"
    server.stop(0)


def _call(target, messages):
    ch = grpc.insecure_channel(target)
    stub = ch.stream_stream(
        "/envoy.service.ext_proc.v3.ExternalProcessor/Process",
        request_serializer=lambda b: b,
        response_deserializer=lambda b: b,
    )
    resp = list(stub(iter(messages)))
    ch.close()
    return resp


def test_extproc_prefix_affinity(extproc_server):
    pods = "pod-a:8000,pod-b:8000"
    prompt = "x" * 300 + " tell me a story"
    msgs = [
        _encode_headers({":path": "/v1/completions",
                         "content-length": "64",
                         "x-gateway-candidate-pods": pods}),
        _encode_body(json.dumps({"prompt": prompt}).encode()),
    ]
    first = _decode_set_headers(_call(extproc_server, msgs)[-1])
    pick1 = first["x-gateway-destination-endpoint"]
    assert pick1 in ("pod-a:8000", "pod-b:8000")
    # the same long prefix must route to the same pod (trie affinity)
    second = _decode_set_headers(_call(extproc_server, msgs)[-1])
    assert second["x-gateway-destination-endpoint"] == pick1


def test_extproc_roundrobin_headers_only():
    server = extproc.serve(port=19103, algorithm="roundrobin", block=False)
    try:
        msgs = [_encode_headers({":path": "/v1/models",
                                 "x-gateway-candidate-pods": "a,b,c"})]
        picks = set()
        for _ in range(3):
            r = _decode_set_headers(_call("127.0.0.1:19103", msgs)[0])
            picks.add(r["x-gateway-destination-endpoint"])
        assert picks == {"a", "b", "c"}
    finally:
        server.stop(0)
