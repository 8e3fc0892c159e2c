"""Runtime protobuf stubs — wire-format compatibility without protoc.

The reference generates its gRPC stubs with protoc
(reference scheduler/Makefile `rpc_stubs`, runtime/protobuf/*.proto);
this image ships the ``google.protobuf`` runtime but no protoc binary,
so the stubs are built AT RUNTIME instead: a proto3-subset parser reads
the committed ``rpc/protos/*.proto`` files, constructs
``FileDescriptorProto``s, and materializes real protobuf message
classes through ``message_factory``.  The classes serialize to the
exact protoc wire format — ``tests/test_pb.py`` proves it by building a
second set of classes from the REFERENCE's own .proto files and
round-tripping bytes between the two.

The live transport's default body codec stays msgpack
(rpc/transport.py); set ``SWQ_RPC_CODEC=proto`` to put these messages
on the wire instead (transport.py resolves the request/response message
for each service method from the parsed ``service`` blocks).

Grammar covered (everything the five schema files use): ``syntax``,
``package``, ``import``, ``enum``, ``message`` with scalar / enum /
message-typed fields, ``repeated``, ``bytes``, ``service`` blocks with
unary rpcs, ``//`` comments.
"""

from __future__ import annotations

import os
import re
from typing import Dict, List, Optional, Tuple

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_SCALAR_TYPES = {
    "double": descriptor_pb2.FieldDescriptorProto.TYPE_DOUBLE,
    "float": descriptor_pb2.FieldDescriptorProto.TYPE_FLOAT,
    "int32": descriptor_pb2.FieldDescriptorProto.TYPE_INT32,
    "int64": descriptor_pb2.FieldDescriptorProto.TYPE_INT64,
    "uint32": descriptor_pb2.FieldDescriptorProto.TYPE_UINT32,
    "uint64": descriptor_pb2.FieldDescriptorProto.TYPE_UINT64,
    "sint32": descriptor_pb2.FieldDescriptorProto.TYPE_SINT32,
    "sint64": descriptor_pb2.FieldDescriptorProto.TYPE_SINT64,
    "bool": descriptor_pb2.FieldDescriptorProto.TYPE_BOOL,
    "string": descriptor_pb2.FieldDescriptorProto.TYPE_STRING,
    "bytes": descriptor_pb2.FieldDescriptorProto.TYPE_BYTES,
}

_TOKEN_RE = re.compile(
    r"//[^\n]*"          # comment
    r"|\"[^\"]*\""       # string literal
    r"|[{}();=]"         # punctuation
    r"|[A-Za-z0-9_.]+",  # identifier / number
)


def _tokenize(text: str) -> List[str]:
    return [t for t in _TOKEN_RE.findall(text) if not t.startswith("//")]


class _Parser:
    def __init__(self, tokens: List[str]):
        self.toks = tokens
        self.i = 0

    def peek(self) -> Optional[str]:
        return self.toks[self.i] if self.i < len(self.toks) else None

    def next(self) -> str:
        t = self.toks[self.i]
        self.i += 1
        return t

    def expect(self, tok: str):
        t = self.next()
        if t != tok:
            raise ValueError(f"expected {tok!r}, got {t!r}")


def parse_proto(text: str) -> dict:
    """-> {package, imports, enums: {name: [(ename, num)]},
    messages: {name: [(fname, num, type, repeated)]},
    services: {sname: {rpc: (req, resp)}}}"""
    p = _Parser(_tokenize(text))
    out = {"package": "", "imports": [], "enums": {}, "messages": {},
           "services": {}}
    while p.peek() is not None:
        t = p.next()
        if t == "syntax":
            p.expect("=")
            p.next()  # "proto3"
            p.expect(";")
        elif t == "package":
            out["package"] = p.next()
            p.expect(";")
        elif t == "import":
            out["imports"].append(p.next().strip('"'))
            p.expect(";")
        elif t == "enum":
            name = p.next()
            p.expect("{")
            values = []
            while p.peek() != "}":
                ename = p.next()
                p.expect("=")
                values.append((ename, int(p.next())))
                p.expect(";")
            p.expect("}")
            out["enums"][name] = values
        elif t == "message":
            name = p.next()
            p.expect("{")
            fields = []
            while p.peek() != "}":
                repeated = False
                ftype = p.next()
                if ftype == "repeated":
                    repeated = True
                    ftype = p.next()
                fname = p.next()
                p.expect("=")
                num = int(p.next())
                p.expect(";")
                fields.append((fname, num, ftype, repeated))
            p.expect("}")
            out["messages"][name] = fields
        elif t == "service":
            sname = p.next()
            p.expect("{")
            rpcs = {}
            while p.peek() != "}":
                p.expect("rpc")
                rname = p.next()
                p.expect("(")
                req = p.next()
                p.expect(")")
                p.expect("returns")
                p.expect("(")
                resp = p.next()
                p.expect(")")
                if p.peek() == "{":
                    p.next()
                    p.expect("}")
                if p.peek() == ";":
                    p.next()
                rpcs[rname] = (req, resp)
            p.expect("}")
            out["services"][sname] = rpcs
        else:
            raise ValueError(f"unexpected token {t!r}")
    return out


class Schema:
    """All messages/enums/services of a proto directory, materialized as
    protobuf classes in a private descriptor pool."""

    def __init__(self, proto_dir: str, pool_name: str):
        self.pool = descriptor_pool.DescriptorPool()
        self.messages: Dict[str, type] = {}
        self.enums: Dict[str, Dict[str, int]] = {}
        self.services: Dict[str, Dict[str, Tuple[str, str]]] = {}

        parsed = {}
        for fname in sorted(os.listdir(proto_dir)):
            if fname.endswith(".proto"):
                parsed[fname] = parse_proto(
                    open(os.path.join(proto_dir, fname)).read()
                )

        # one namespace across files (they cross-import); package from
        # any file that declares one (wire bytes never carry it)
        package = next(
            (d["package"] for d in parsed.values() if d["package"]),
            "swq_pb_" + pool_name,
        )
        all_enums = {}
        all_messages = {}
        for d in parsed.values():
            all_enums.update(d["enums"])
            all_messages.update(d["messages"])
            self.services.update(d["services"])

        fdp = descriptor_pb2.FileDescriptorProto()
        fdp.name = pool_name + ".proto"
        fdp.package = package
        fdp.syntax = "proto3"
        for ename, values in all_enums.items():
            e = fdp.enum_type.add()
            e.name = ename
            for vname, vnum in values:
                v = e.value.add()
                v.name = vname
                v.number = vnum
        for mname, fields in all_messages.items():
            m = fdp.message_type.add()
            m.name = mname
            for fname_, num, ftype, repeated in fields:
                f = m.field.add()
                f.name = fname_
                f.number = num
                f.label = (
                    descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
                    if repeated
                    else descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
                )
                if ftype in _SCALAR_TYPES:
                    f.type = _SCALAR_TYPES[ftype]
                elif ftype in all_enums:
                    f.type = descriptor_pb2.FieldDescriptorProto.TYPE_ENUM
                    f.type_name = f".{package}.{ftype}"
                elif ftype in all_messages:
                    f.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
                    f.type_name = f".{package}.{ftype}"
                else:
                    raise ValueError(f"unknown field type {ftype!r}")

        file_desc = self.pool.Add(fdp)
        for mname in all_messages:
            desc = self.pool.FindMessageTypeByName(f"{package}.{mname}")
            self.messages[mname] = message_factory.GetMessageClass(desc)
        for ename, values in all_enums.items():
            self.enums[ename] = dict(values)
        self.package = package
        self._file_desc = file_desc

    # -- dict <-> message (the services speak dicts keyed by field name)

    def to_message(self, mname: str, payload: Optional[dict]):
        msg = self.messages[mname]()
        for f in msg.DESCRIPTOR.fields:
            if not payload or f.name not in payload or payload[f.name] is None:
                continue
            v = payload[f.name]
            if f.is_repeated:
                if f.type == f.TYPE_MESSAGE:
                    for item in v:
                        self._fill(getattr(msg, f.name).add(), item)
                else:
                    getattr(msg, f.name).extend(v)
            elif f.type == f.TYPE_MESSAGE:
                self._fill(getattr(msg, f.name), v)
            else:
                setattr(msg, f.name, v)
        return msg

    def _fill(self, sub, d: dict):
        for f in sub.DESCRIPTOR.fields:
            if f.name in d and d[f.name] is not None:
                if f.is_repeated:
                    getattr(sub, f.name).extend(d[f.name])
                else:
                    setattr(sub, f.name, d[f.name])

    def to_dict(self, msg) -> dict:
        out = {}
        for f in msg.DESCRIPTOR.fields:
            v = getattr(msg, f.name)
            if f.is_repeated:
                if f.type == f.TYPE_MESSAGE:
                    out[f.name] = [self.to_dict(x) for x in v]
                else:
                    out[f.name] = list(v)
            elif f.type == f.TYPE_MESSAGE:
                out[f.name] = self.to_dict(v)
            else:
                out[f.name] = v
        return out

    def method_types(self, service: str, method: str) -> Tuple[str, str]:
        return self.services[service][method]


_OUR_SCHEMA: Optional[Schema] = None


def our_schema() -> Schema:
    global _OUR_SCHEMA
    if _OUR_SCHEMA is None:
        _OUR_SCHEMA = Schema(
            os.path.join(os.path.dirname(__file__), "protos"), "shockwave_amd"
        )
    return _OUR_SCHEMA
