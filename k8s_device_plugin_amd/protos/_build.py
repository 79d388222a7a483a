"""Tiny descriptor builder: protobuf message classes without protoc.

The build image has the protobuf runtime but no protoc / grpcio-tools, so the
kubelet DevicePlugin v1beta1 schema and the metricssvc schema are authored as
FileDescriptorProtos in Python and turned into real message classes via
message_factory.  Field names/numbers/types are transcribed 1:1 from the
reference protos, so the wire format is identical to the kubelet ABI
(reference: vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto and
internal/pkg/exporter/metricssvc/metricssvc.pb.go).
"""

from __future__ import annotations

from typing import Dict, Sequence, Tuple

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

F = descriptor_pb2.FieldDescriptorProto

_SCALAR_TYPES = {
    "string": F.TYPE_STRING,
    "bool": F.TYPE_BOOL,
    "int32": F.TYPE_INT32,
    "int64": F.TYPE_INT64,
    "uint32": F.TYPE_UINT32,
    "uint64": F.TYPE_UINT64,
    "bytes": F.TYPE_BYTES,
}

# Field spec: (name, number, type[, "repeated"]); type is a scalar name
# above, "msg:<fully.qualified.Type>", or "map<string,string>"
class FileBuilder:
    def __init__(self, name: str, package: str, dependencies: Sequence[str] = ()):
        self.fdp = descriptor_pb2.FileDescriptorProto()
        self.fdp.name = name
        self.fdp.package = package
        self.fdp.syntax = "proto3"
        for dep in dependencies:
            self.fdp.dependency.append(dep)

    def message(self, name: str, fields: Sequence[Tuple] = ()) -> None:
        """fields: (name, number, type) or (name, number, type, 'repeated')."""
        msg = self.fdp.message_type.add()
        msg.name = name
        for spec in fields:
            fname, number, ftype = spec[0], spec[1], spec[2]
            repeated = len(spec) > 3 and spec[3] == "repeated"
            fld = msg.field.add()
            fld.name = fname
            fld.number = number
            fld.json_name = fname
            if ftype.startswith("map<"):
                # map<string,string> only (all the reference uses)
                entry_name = _map_entry_name(fname)
                entry = msg.nested_type.add()
                entry.name = entry_name
                entry.options.map_entry = True
                for kname, knum in (("key", 1), ("value", 2)):
                    kf = entry.field.add()
                    kf.name = kname
                    kf.number = knum
                    kf.type = F.TYPE_STRING
                    kf.label = F.LABEL_OPTIONAL
                    kf.json_name = kname
                fld.type = F.TYPE_MESSAGE
                fld.label = F.LABEL_REPEATED
                fld.type_name = f".{self.fdp.package}.{name}.{entry_name}"
            elif ftype.startswith("msg:"):
                fld.type = F.TYPE_MESSAGE
                fld.label = F.LABEL_REPEATED if repeated else F.LABEL_OPTIONAL
                fld.type_name = "." + ftype[4:]
            else:
                fld.type = _SCALAR_TYPES[ftype]
                fld.label = F.LABEL_REPEATED if repeated else F.LABEL_OPTIONAL

    def build(self, pool: descriptor_pool.DescriptorPool) -> Dict[str, type]:
        """Add to pool; return {message_name: message_class}."""
        try:
            fd = pool.Add(self.fdp)
        except Exception:
            # already registered (module re-import with a shared pool)
            fd = pool.FindFileByName(self.fdp.name)
        out: Dict[str, type] = {}
        for mname, mdesc in fd.message_types_by_name.items():
            out[mname] = message_factory.GetMessageClass(mdesc)
        return out


def _map_entry_name(field_name: str) -> str:
    return "".join(p.capitalize() for p in field_name.split("_")) + "Entry"


def make_pool_with_wellknown() -> descriptor_pool.DescriptorPool:
    """A private pool pre-loaded with google/protobuf/empty.proto."""
    from google.protobuf import empty_pb2

    pool = descriptor_pool.DescriptorPool()
    pool.Add(
        descriptor_pb2.FileDescriptorProto.FromString(
            empty_pb2.DESCRIPTOR.serialized_pb
        )
    )
    return pool
