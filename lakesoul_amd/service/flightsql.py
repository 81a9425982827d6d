"""Arrow Flight SQL protocol envelopes (reference:
rust/lakesoul-flight/src/flight_sql_service.rs:218-1085).

Flight SQL rides plain Flight: a client packs protobuf commands into a
``google.protobuf.Any`` and puts them in FlightDescriptor.cmd /
Ticket.ticket. This module hand-rolls exactly those wire messages
(FlightSql.proto field numbers; same approach as io/substrait.py — no
generated stubs) so standard Flight SQL clients' GetFlightInfo/DoGet/
DoPut flows are wire-compatible:

- CommandStatementQuery{query=1}        -> FlightInfo + TicketStatementQuery
- TicketStatementQuery{statement_handle=1}
- CommandStatementUpdate{query=1}       -> DoPut, DoPutUpdateResult{record_count=1}
- CommandGetCatalogs{} / CommandGetDbSchemas{catalog=1, pattern=2}
- CommandGetTables{catalog=1, schema_pattern=2, table_pattern=3,
                   table_types=4, include_schema=5}
"""

from __future__ import annotations

from typing import Optional, Tuple

_PREFIX = "type.googleapis.com/arrow.flight.protocol.sql."


# ---- minimal protobuf wire helpers (shared shapes with io/substrait) ---- #

def _w_varint(v: int) -> bytes:
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _w_len(fn: int, payload: bytes) -> bytes:
    return _w_varint((fn << 3) | 2) + _w_varint(len(payload)) + payload


def _w_vint(fn: int, v: int) -> bytes:
    return _w_varint((fn << 3) | 0) + _w_varint(v)


def _fields(b: bytes):
    i = 0
    n = len(b)
    while i < n:
        tag = 0
        s = 0
        while True:
            x = b[i]
            i += 1
            tag |= (x & 0x7F) << s
            if not x & 0x80:
                break
            s += 7
        fn, wt = tag >> 3, tag & 7
        if wt == 0:
            v = 0
            s = 0
            while True:
                x = b[i]
                i += 1
                v |= (x & 0x7F) << s
                if not x & 0x80:
                    break
                s += 7
            yield fn, v
        elif wt == 2:
            ln = 0
            s = 0
            while True:
                x = b[i]
                i += 1
                ln |= (x & 0x7F) << s
                if not x & 0x80:
                    break
                s += 7
            yield fn, b[i:i + ln]
            i += ln
        elif wt == 5:
            yield fn, b[i:i + 4]
            i += 4
        elif wt == 1:
            yield fn, b[i:i + 8]
            i += 8
        else:
            raise ValueError(f"wire type {wt}")


# ---- google.protobuf.Any ---- #

def pack_any(short_name: str, payload: bytes) -> bytes:
    return _w_len(1, (_PREFIX + short_name).encode()) + _w_len(2, payload)


def unpack_any(buf: bytes) -> Tuple[Optional[str], bytes]:
    """(short type name if a flight-sql type, payload)."""
    url, payload = "", b""
    for fn, v in _fields(buf):
        if fn == 1 and isinstance(v, (bytes, bytearray)):
            url = bytes(v).decode()
        elif fn == 2 and isinstance(v, (bytes, bytearray)):
            payload = bytes(v)
    if url.startswith(_PREFIX):
        return url[len(_PREFIX):], payload
    return None, payload


# ---- commands ---- #

def cmd_statement_query(query: str) -> bytes:
    return pack_any("CommandStatementQuery", _w_len(1, query.encode()))


def cmd_statement_update(query: str) -> bytes:
    return pack_any("CommandStatementUpdate", _w_len(1, query.encode()))


def cmd_get_catalogs() -> bytes:
    return pack_any("CommandGetCatalogs", b"")


def cmd_get_db_schemas(catalog: str = "") -> bytes:
    body = _w_len(1, catalog.encode()) if catalog else b""
    return pack_any("CommandGetDbSchemas", body)


def cmd_get_tables(include_schema: bool = False) -> bytes:
    body = b""
    if include_schema:
        body += _w_vint(5, 1)
    return pack_any("CommandGetTables", body)


def ticket_statement_query(handle: bytes) -> bytes:
    return pack_any("TicketStatementQuery", _w_len(1, handle))


def parse_string_field(payload: bytes, field: int = 1) -> str:
    for fn, v in _fields(payload):
        if fn == field and isinstance(v, (bytes, bytearray)):
            return bytes(v).decode()
    return ""


def parse_bytes_field(payload: bytes, field: int = 1) -> bytes:
    for fn, v in _fields(payload):
        if fn == field and isinstance(v, (bytes, bytearray)):
            return bytes(v)
    return b""


def parse_bool_field(payload: bytes, field: int) -> bool:
    for fn, v in _fields(payload):
        if fn == field and isinstance(v, int):
            return bool(v)
    return False


def do_put_update_result(record_count: int) -> bytes:
    return _w_vint(1, record_count)


def parse_do_put_update_result(buf: bytes) -> int:
    for fn, v in _fields(buf):
        if fn == 1 and isinstance(v, int):
            return v
    return -1
