"""Runtime proto3 compiler for the service wire contract.

The image ships ``grpcio`` and ``google.protobuf`` but no ``protoc`` /
``grpcio-tools``, so the pb2 modules the reference generates offline
(ols_core/proto/*_pb2*.py) cannot be produced the usual way.  This
module compiles the subset of proto3 those service definitions use —
top-level messages, enums, scalar / repeated fields, message-typed
fields, ``google.protobuf.Empty`` imports, services with unary RPCs —
directly into live protobuf message classes via
``descriptor_pb2.FileDescriptorProto`` + ``message_factory``, giving
REAL protobuf wire compatibility: a client built from the same .proto
files with stock protoc interoperates byte-for-byte.

Supported syntax (all six service files fit):
  syntax/package/option/import statements, ``enum`` with numbered
  values, ``message`` with scalar|enum|message fields (optionally
  ``repeated``), ``service`` with ``rpc Name(Req) returns (Resp);``.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_SCALARS = {
    "double": descriptor_pb2.FieldDescriptorProto.TYPE_DOUBLE,
    "float": descriptor_pb2.FieldDescriptorProto.TYPE_FLOAT,
    "int64": descriptor_pb2.FieldDescriptorProto.TYPE_INT64,
    "uint64": descriptor_pb2.FieldDescriptorProto.TYPE_UINT64,
    "int32": descriptor_pb2.FieldDescriptorProto.TYPE_INT32,
    "uint32": descriptor_pb2.FieldDescriptorProto.TYPE_UINT32,
    "bool": descriptor_pb2.FieldDescriptorProto.TYPE_BOOL,
    "string": descriptor_pb2.FieldDescriptorProto.TYPE_STRING,
    "bytes": descriptor_pb2.FieldDescriptorProto.TYPE_BYTES,
}

_TOKEN = re.compile(
    r"//[^\n]*|/\*.*?\*/|\"(?:[^\"\\]|\\.)*\"|'(?:[^'\\]|\\.)*'"
    r"|[A-Za-z_][A-Za-z0-9_.]*|-?\d+|[{}();=\[\]<>,]", re.S)


def _tokens(text: str) -> List[str]:
    return [t for t in _TOKEN.findall(text)
            if not (t.startswith("//") or t.startswith("/*"))]


@dataclass
class RpcDef:
    name: str
    request: str
    response: str


@dataclass
class ServiceDef:
    name: str
    rpcs: List[RpcDef] = field(default_factory=list)


class ProtoFile:
    """One parsed .proto file, compiled into a descriptor pool."""

    def __init__(self, text: str, name: str = "mini.proto",
                 pool: Optional[descriptor_pool.DescriptorPool] = None):
        self.pool = pool or descriptor_pool.DescriptorPool()
        self.package = ""
        self.services: Dict[str, ServiceDef] = {}
        self._fdp = descriptor_pb2.FileDescriptorProto()
        self._fdp.name = name
        self._fdp.syntax = "proto3"
        self._enum_names: set = set()
        self._parse(_tokens(text))
        self._fdp.package = self.package
        # register the well-known Empty into OUR pool (a private pool
        # does not inherit the default pool's well-known types)
        if any(d == "google/protobuf/empty.proto"
               for d in self._fdp.dependency):
            try:
                self.pool.Add(_empty_fdp())
            except Exception:
                pass  # already present
        self._fd = self.pool.Add(self._fdp)
        self._classes: Dict[str, type] = {}

    # -- public ---------------------------------------------------------
    def message_class(self, name: str) -> type:
        """Live message class for a (possibly package-qualified) name."""
        cls = self._classes.get(name)
        if cls is None:
            full = self._qualify(name)
            desc = self.pool.FindMessageTypeByName(full)
            cls = message_factory.GetMessageClass(desc)
            self._classes[name] = cls
        return cls

    def _qualify(self, name: str) -> str:
        if "." in name:
            return name
        return f"{self.package}.{name}" if self.package else name

    # -- parsing --------------------------------------------------------
    def _parse(self, toks: List[str]) -> None:
        i = 0
        n = len(toks)
        while i < n:
            t = toks[i]
            if t == "syntax":
                i = self._skip_to(toks, i, ";") + 1
            elif t == "package":
                self.package = toks[i + 1]
                i = self._skip_to(toks, i, ";") + 1
            elif t == "option":
                i = self._skip_to(toks, i, ";") + 1
            elif t == "import":
                dep = toks[i + 1].strip("\"'")
                self._fdp.dependency.append(dep)
                i = self._skip_to(toks, i, ";") + 1
            elif t == "enum":
                i = self._parse_enum(toks, i)
            elif t == "message":
                i = self._parse_message(toks, i)
            elif t == "service":
                i = self._parse_service(toks, i)
            else:
                i += 1
        # enum-vs-message resolution must wait until every enum in the
        # file has been seen (enums may be declared after their users)
        for md in self._fdp.message_type:
            self._pending_fixup(md)

    @staticmethod
    def _skip_to(toks: List[str], i: int, stop: str) -> int:
        while i < len(toks) and toks[i] != stop:
            i += 1
        return i

    def _parse_enum(self, toks: List[str], i: int) -> int:
        name = toks[i + 1]
        self._enum_names.add(name)
        ed = self._fdp.enum_type.add()
        ed.name = name
        i += 3  # enum NAME {
        while toks[i] != "}":
            vname = toks[i]
            assert toks[i + 1] == "=", f"enum {name}: expected '='"
            v = ed.value.add()
            v.name = vname
            v.number = int(toks[i + 2])
            i = self._skip_to(toks, i, ";") + 1
        return i + 1

    def _parse_message(self, toks: List[str], i: int) -> int:
        name = toks[i + 1]
        md = self._fdp.message_type.add()
        md.name = name
        i += 3  # message NAME {
        fnum = 0
        while toks[i] != "}":
            repeated = False
            if toks[i] == "repeated":
                repeated = True
                i += 1
            ftype = toks[i]
            fname = toks[i + 1]
            assert toks[i + 2] == "=", \
                f"message {name}.{fname}: expected '=' (got {toks[i+2]!r})"
            fnum = int(toks[i + 3])
            f = md.field.add()
            f.name = fname
            f.number = fnum
            f.label = (descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
                       if repeated
                       else descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL)
            if ftype in _SCALARS:
                f.type = _SCALARS[ftype]
            elif ftype == "google.protobuf.Empty":
                f.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
                f.type_name = ".google.protobuf.Empty"
            else:
                # enum vs message resolved after parsing (enums may be
                # declared later in the file) — leave type unset and fix
                # in the second pass; the pool resolves either kind
                f.type_name = ftype          # patched to full name below
            i = self._skip_to(toks, i, ";") + 1
        return i + 1

    def _pending_fixup(self, md) -> None:
        for f in md.field:
            if not f.HasField("type") and f.type_name and \
                    not f.type_name.startswith("."):
                local = f.type_name
                pkg = f".{self.package}." if self.package else "."
                if local in self._enum_names:
                    f.type = descriptor_pb2.FieldDescriptorProto.TYPE_ENUM
                else:
                    f.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
                f.type_name = pkg + local

    def _parse_service(self, toks: List[str], i: int) -> int:
        name = toks[i + 1]
        svc = ServiceDef(name=name)
        sd = self._fdp.service.add()
        sd.name = name
        i += 3  # service NAME {
        while toks[i] != "}":
            assert toks[i] == "rpc", f"service {name}: expected rpc"
            rpc_name = toks[i + 1]
            assert toks[i + 2] == "("
            req = toks[i + 3]
            assert toks[i + 4] == ")" and toks[i + 5] == "returns"
            assert toks[i + 6] == "("
            resp = toks[i + 7]
            svc.rpcs.append(RpcDef(rpc_name, req, resp))
            m = sd.method.add()
            m.name = rpc_name
            m.input_type = self._type_ref(req)
            m.output_type = self._type_ref(resp)
            i = self._skip_to(toks, i, ";") + 1
        self.services[name] = svc
        return i + 1

    def _type_ref(self, name: str) -> str:
        if name == "google.protobuf.Empty":
            return ".google.protobuf.Empty"
        return (f".{self.package}.{name}") if self.package else f".{name}"


def _empty_fdp() -> descriptor_pb2.FileDescriptorProto:
    """google/protobuf/empty.proto as a FileDescriptorProto (copied from
    the installed well-known type's serialized descriptor)."""
    from google.protobuf import empty_pb2
    fdp = descriptor_pb2.FileDescriptorProto()
    empty_pb2.DESCRIPTOR.CopyToProto(fdp)
    return fdp


def load_proto(path: str,
               pool: Optional[descriptor_pool.DescriptorPool] = None
               ) -> ProtoFile:
    with open(path) as f:
        text = f.read()
    import os
    return ProtoFile(text, name=os.path.basename(path), pool=pool)
