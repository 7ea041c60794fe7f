"""Protobuf wire codec for the Envoy external-processor protocol subset.

The reference gateway is an Envoy ext_proc plugin
(envoy/service/ext_proc/v3/external_processor.proto, consumed at
reference pkg/gateway/gateway.go:77-138). protoc is not available in this
image, so the exact wire format of the messages the gateway exchanges is
implemented directly: dataclasses + proto3 varint/length-delimited
encoding. Field numbers below are the ones from the envoy proto files
(cited per message); any ext_proc client generated from those protos
interoperates on the wire.

Covered messages:
  ProcessingRequest  (oneof request_headers=2 / response_headers=3 /
                      request_body=4 / response_body=5, observability=10)
  ProcessingResponse (oneof request_headers=1 / response_headers=2 /
                      request_body=3 / response_body=4 /
                      immediate_response=7)
  HttpHeaders{header_map=1, end_of_stream=3}, HttpBody{body=1, eos=2}
  CommonResponse{status=1, header_mutation=2, clear_route_cache=5}
  HeaderMutation{set_headers=1, remove_headers=2}
  HeaderValueOption{header=1}, HeaderValue{key=1, value=2, raw_value=3}
  ImmediateResponse{status{code=1}=1, headers=2, body=3}
"""

from __future__ import annotations

from dataclasses import dataclass, field


# ---------------- low-level proto3 wire helpers ----------------
def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _tag(fieldnum: int, wire: int) -> bytes:
    return _varint((fieldnum << 3) | wire)


def _len_field(fieldnum: int, payload: bytes) -> bytes:
    return _tag(fieldnum, 2) + _varint(len(payload)) + payload


def _bool_field(fieldnum: int, v: bool) -> bytes:
    return (_tag(fieldnum, 0) + b"\x01") if v else b""


def _varint_field(fieldnum: int, v: int) -> bytes:
    return (_tag(fieldnum, 0) + _varint(v)) if v else b""


class _Reader:
    def __init__(self, data: bytes):
        self.data = data
        self.pos = 0

    def eof(self) -> bool:
        return self.pos >= len(self.data)

    def varint(self) -> int:
        shift = 0
        out = 0
        while True:
            b = self.data[self.pos]
            self.pos += 1
            out |= (b & 0x7F) << shift
            if not b & 0x80:
                return out
            shift += 7

    def tag(self) -> tuple[int, int]:
        t = self.varint()
        return t >> 3, t & 7

    def bytes_(self) -> bytes:
        n = self.varint()
        out = self.data[self.pos:self.pos + n]
        self.pos += n
        return out

    def skip(self, wire: int) -> None:
        if wire == 0:
            self.varint()
        elif wire == 1:
            self.pos += 8
        elif wire == 2:
            self.bytes_()
        elif wire == 5:
            self.pos += 4
        else:
            raise ValueError(f"unsupported wire type {wire}")


# ---------------- message dataclasses ----------------
@dataclass
class HeaderValue:  # envoy.config.core.v3.HeaderValue
    key: str = ""
    value: str = ""
    raw_value: bytes = b""

    def encode(self) -> bytes:
        out = b""
        if self.key:
            out += _len_field(1, self.key.encode())
        if self.value:
            out += _len_field(2, self.value.encode())
        if self.raw_value:
            out += _len_field(3, self.raw_value)
        return out

    @classmethod
    def decode(cls, data: bytes) -> "HeaderValue":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 1:
                out.key = r.bytes_().decode()
            elif f == 2:
                out.value = r.bytes_().decode()
            elif f == 3:
                out.raw_value = r.bytes_()
            else:
                r.skip(w)
        return out

    def text(self) -> str:
        """Envoy sends header values in raw_value; generated clients may
        set either."""
        return self.value or self.raw_value.decode("utf-8", "replace")


@dataclass
class HeaderMap:  # envoy.config.core.v3.HeaderMap {headers=1}
    headers: list[HeaderValue] = field(default_factory=list)

    def encode(self) -> bytes:
        return b"".join(_len_field(1, h.encode()) for h in self.headers)

    @classmethod
    def decode(cls, data: bytes) -> "HeaderMap":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 1:
                out.headers.append(HeaderValue.decode(r.bytes_()))
            else:
                r.skip(w)
        return out


@dataclass
class HttpHeaders:  # ext_proc v3 HttpHeaders {headers=1, end_of_stream=3}
    headers: HeaderMap = field(default_factory=HeaderMap)
    end_of_stream: bool = False

    def encode(self) -> bytes:
        return (_len_field(1, self.headers.encode())
                + _bool_field(3, self.end_of_stream))

    @classmethod
    def decode(cls, data: bytes) -> "HttpHeaders":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 1:
                out.headers = HeaderMap.decode(r.bytes_())
            elif f == 3:
                out.end_of_stream = bool(r.varint())
            else:
                r.skip(w)
        return out

    def get(self, key: str) -> str | None:
        for h in self.headers.headers:
            if h.key.lower() == key.lower():
                return h.text()
        return None


@dataclass
class HttpBody:  # ext_proc v3 HttpBody {body=1, end_of_stream=2}
    body: bytes = b""
    end_of_stream: bool = False

    def encode(self) -> bytes:
        out = b""
        if self.body:
            out += _len_field(1, self.body)
        out += _bool_field(2, self.end_of_stream)
        return out

    @classmethod
    def decode(cls, data: bytes) -> "HttpBody":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 1:
                out.body = r.bytes_()
            elif f == 2:
                out.end_of_stream = bool(r.varint())
            else:
                r.skip(w)
        return out


@dataclass
class ProcessingRequest:
    """ext_proc v3 ProcessingRequest: oneof request {request_headers=2,
    response_headers=3, request_body=4, response_body=5}."""

    request_headers: HttpHeaders | None = None
    response_headers: HttpHeaders | None = None
    request_body: HttpBody | None = None
    response_body: HttpBody | None = None

    def encode(self) -> bytes:
        if self.request_headers is not None:
            return _len_field(2, self.request_headers.encode())
        if self.response_headers is not None:
            return _len_field(3, self.response_headers.encode())
        if self.request_body is not None:
            return _len_field(4, self.request_body.encode())
        if self.response_body is not None:
            return _len_field(5, self.response_body.encode())
        return b""

    @classmethod
    def decode(cls, data: bytes) -> "ProcessingRequest":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 2:
                out.request_headers = HttpHeaders.decode(r.bytes_())
            elif f == 3:
                out.response_headers = HttpHeaders.decode(r.bytes_())
            elif f == 4:
                out.request_body = HttpBody.decode(r.bytes_())
            elif f == 5:
                out.response_body = HttpBody.decode(r.bytes_())
            else:
                r.skip(w)
        return out


@dataclass
class HeaderMutation:  # CommonResponse.HeaderMutation {set=1, remove=2}
    set_headers: list[HeaderValue] = field(default_factory=list)
    remove_headers: list[str] = field(default_factory=list)

    def encode(self) -> bytes:
        out = b""
        for h in self.set_headers:
            # HeaderValueOption{header=1}
            out += _len_field(1, _len_field(1, h.encode()))
        for k in self.remove_headers:
            out += _len_field(2, k.encode())
        return out

    @classmethod
    def decode(cls, data: bytes) -> "HeaderMutation":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 1:
                opt = _Reader(r.bytes_())
                while not opt.eof():
                    f2, w2 = opt.tag()
                    if f2 == 1:
                        out.set_headers.append(HeaderValue.decode(opt.bytes_()))
                    else:
                        opt.skip(w2)
            elif f == 2:
                out.remove_headers.append(r.bytes_().decode())
            else:
                r.skip(w)
        return out


@dataclass
class CommonResponse:
    """ext_proc v3 CommonResponse {status=1 (0=CONTINUE),
    header_mutation=2, clear_route_cache=5}."""

    header_mutation: HeaderMutation | None = None
    clear_route_cache: bool = False

    def encode(self) -> bytes:
        out = b""
        if self.header_mutation is not None:
            out += _len_field(2, self.header_mutation.encode())
        out += _bool_field(5, self.clear_route_cache)
        return out

    @classmethod
    def decode(cls, data: bytes) -> "CommonResponse":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 2:
                out.header_mutation = HeaderMutation.decode(r.bytes_())
            elif f == 5:
                out.clear_route_cache = bool(r.varint())
            else:
                r.skip(w)
        return out


@dataclass
class ImmediateResponse:
    """ext_proc v3 ImmediateResponse {status=1 (HttpStatus{code=1}),
    headers=2, body=3} — terminates the HTTP request at Envoy with this
    status/body (the gateway's 401/400/429 path, reference util.go:40-77)."""

    status_code: int = 200
    headers: HeaderMutation | None = None
    body: bytes = b""

    def encode(self) -> bytes:
        out = _len_field(1, _varint_field(1, self.status_code))
        if self.headers is not None:
            out += _len_field(2, self.headers.encode())
        if self.body:
            out += _len_field(3, self.body)
        return out

    @classmethod
    def decode(cls, data: bytes) -> "ImmediateResponse":
        r, out = _Reader(data), cls()
        while not r.eof():
            f, w = r.tag()
            if f == 1:
                st = _Reader(r.bytes_())
                while not st.eof():
                    f2, w2 = st.tag()
                    if f2 == 1:
                        out.status_code = st.varint()
                    else:
                        st.skip(w2)
            elif f == 2:
                out.headers = HeaderMutation.decode(r.bytes_())
            elif f == 3:
                out.body = r.bytes_()
            else:
                r.skip(w)
        return out


@dataclass
class ProcessingResponse:
    """ext_proc v3 ProcessingResponse: oneof response {request_headers=1,
    response_headers=2, request_body=3, response_body=4,
    immediate_response=7}; headers/body responses wrap CommonResponse at
    field 1 (HeadersResponse{response=1} / BodyResponse{response=1})."""

    request_headers: CommonResponse | None = None
    response_headers: CommonResponse | None = None
    request_body: CommonResponse | None = None
    response_body: CommonResponse | None = None
    immediate_response: ImmediateResponse | None = None

    def encode(self) -> bytes:
        if self.request_headers is not None:
            return _len_field(1, _len_field(1, self.request_headers.encode()))
        if self.response_headers is not None:
            return _len_field(2, _len_field(1, self.response_headers.encode()))
        if self.request_body is not None:
            return _len_field(3, _len_field(1, self.request_body.encode()))
        if self.response_body is not None:
            return _len_field(4, _len_field(1, self.response_body.encode()))
        if self.immediate_response is not None:
            return _len_field(7, self.immediate_response.encode())
        return b""

    @classmethod
    def decode(cls, data: bytes) -> "ProcessingResponse":
        r, out = _Reader(data), cls()

        def common(payload: bytes) -> CommonResponse:
            rr = _Reader(payload)
            while not rr.eof():
                f2, w2 = rr.tag()
                if f2 == 1:
                    return CommonResponse.decode(rr.bytes_())
                rr.skip(w2)
            return CommonResponse()

        while not r.eof():
            f, w = r.tag()
            if f == 1:
                out.request_headers = common(r.bytes_())
            elif f == 2:
                out.response_headers = common(r.bytes_())
            elif f == 3:
                out.request_body = common(r.bytes_())
            elif f == 4:
                out.response_body = common(r.bytes_())
            elif f == 7:
                out.immediate_response = ImmediateResponse.decode(r.bytes_())
            else:
                r.skip(w)
        return out
