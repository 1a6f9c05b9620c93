"""FakePodResources — kubelet PodResources gRPC server double.

Serves ``v1.PodResourcesLister/List`` on a unix socket using grpcio with
messages built through the *real* protobuf runtime (descriptors constructed
at import time), so the native client's hand-rolled HTTP/2 + protobuf
decoding (native/exporter/podresources.cpp) is validated against an
independent implementation of both wire formats.
"""

from __future__ import annotations

from concurrent import futures

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory


def _build_messages():
    pool = descriptor_pool.DescriptorPool()
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "podresources_fixture.proto"
    f.package = "v1"
    f.syntax = "proto3"

    def add_msg(name):
        m = f.message_type.add()
        m.name = name
        return m

    def add_field(msg, name, number, ftype, label, type_name=None):
        fld = msg.field.add()
        fld.name = name
        fld.number = number
        fld.type = ftype
        fld.label = label
        if type_name:
            fld.type_name = type_name

    T = descriptor_pb2.FieldDescriptorProto
    req = add_msg("ListPodResourcesRequest")
    del req  # empty message

    cd = add_msg("ContainerDevices")
    add_field(cd, "resource_name", 1, T.TYPE_STRING, T.LABEL_OPTIONAL)
    add_field(cd, "device_ids", 2, T.TYPE_STRING, T.LABEL_REPEATED)

    cr = add_msg("ContainerResources")
    add_field(cr, "name", 1, T.TYPE_STRING, T.LABEL_OPTIONAL)
    add_field(cr, "devices", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".v1.ContainerDevices")

    pr = add_msg("PodResources")
    add_field(pr, "name", 1, T.TYPE_STRING, T.LABEL_OPTIONAL)
    add_field(pr, "namespace", 2, T.TYPE_STRING, T.LABEL_OPTIONAL)
    add_field(pr, "containers", 3, T.TYPE_MESSAGE, T.LABEL_REPEATED,
              ".v1.ContainerResources")

    resp = add_msg("ListPodResourcesResponse")
    add_field(resp, "pod_resources", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED,
              ".v1.PodResources")

    pool.Add(f)
    get = lambda name: message_factory.GetMessageClass(pool.FindMessageTypeByName(name))
    return {
        "Response": get("v1.ListPodResourcesResponse"),
    }


_MSGS = _build_messages()


class FakePodResources:
    """Start with a list of entries:
    [{"pod": ..., "namespace": ..., "containers":
        [{"name": ..., "devices": [{"resource_name": ..., "device_ids": [...]}]}]}]
    """

    def __init__(self, socket_path: str, entries: list[dict]):
        self.socket_path = socket_path
        self.entries = entries
        self.calls = 0

        fixture = self

        def list_handler(request_bytes, context):
            fixture.calls += 1
            resp = _MSGS["Response"]()
            for e in fixture.entries:
                pr = resp.pod_resources.add()
                pr.name = e["pod"]
                setattr(pr, "namespace", e["namespace"])
                for c in e.get("containers", []):
                    cr = pr.containers.add()
                    cr.name = c["name"]
                    for d in c.get("devices", []):
                        cd = cr.devices.add()
                        cd.resource_name = d["resource_name"]
                        cd.device_ids.extend(d["device_ids"])
            return resp.SerializeToString()

        handler = grpc.method_handlers_generic_handler(
            "v1.PodResourcesLister",
            {
                "List": grpc.unary_unary_rpc_method_handler(
                    list_handler,
                    request_deserializer=lambda b: b,
                    response_serializer=lambda b: b,
                )
            },
        )
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
        self._server.add_generic_rpc_handlers((handler,))
        self._server.add_insecure_port(f"unix:{socket_path}")

    def start(self):
        self._server.start()
        return self

    def stop(self):
        self._server.stop(grace=None)

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
