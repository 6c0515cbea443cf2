"""gRPC serving path (role of the reference's
vllm/entrypoints/grpc_server.py:56 and rust/proto/inference.proto).

The message classes are built at runtime from a FileDescriptorProto that
mirrors inference.proto in this directory (no protoc available offline);
the wire format is standard protobuf, so clients codegen'd from the
.proto interoperate. The service runs on grpc.aio over the same AsyncLLM
the HTTP server uses.
"""

from __future__ import annotations

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_PKG = "vllm_amd.inference"


def _build_messages():
    """Construct the message classes from a descriptor equivalent to
    inference.proto."""
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "vllm_amd/inference.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"

    T = descriptor_pb2.FieldDescriptorProto

    def msg(name, fields):
        m = fdp.message_type.add()
        m.name = name
        for num, (fname, ftype, label, tname) in enumerate(fields, 1):
            f = m.field.add()
            f.name = fname
            f.number = num
            f.type = ftype
            f.label = label
            if tname:
                f.type_name = f".{_PKG}.{tname}"

    OPT = T.LABEL_OPTIONAL
    REP = T.LABEL_REPEATED
    msg("SamplingOptions", [
        ("temperature", T.TYPE_FLOAT, OPT, None),
        ("top_p", T.TYPE_FLOAT, OPT, None),
        ("top_k", T.TYPE_INT32, OPT, None),
        ("max_tokens", T.TYPE_INT32, OPT, None),
        ("seed", T.TYPE_UINT64, OPT, None),
        ("has_seed", T.TYPE_BOOL, OPT, None),
        ("ignore_eos", T.TYPE_BOOL, OPT, None),
        ("stop", T.TYPE_STRING, REP, None),
    ])
    msg("GenerateRequest", [
        ("request_id", T.TYPE_STRING, OPT, None),
        ("prompt", T.TYPE_STRING, OPT, None),
        ("prompt_token_ids", T.TYPE_UINT32, REP, None),
        ("sampling", T.TYPE_MESSAGE, OPT, "SamplingOptions"),
        ("lora", T.TYPE_STRING, OPT, None),
    ])
    msg("GenerateChunk", [
        ("token_ids", T.TYPE_UINT32, REP, None),
        ("text", T.TYPE_STRING, OPT, None),
        ("finish_reason", T.TYPE_STRING, OPT, None),
        ("prompt_tokens", T.TYPE_UINT32, OPT, None),
    ])
    msg("EmbedRequest", [
        ("prompt", T.TYPE_STRING, OPT, None),
        ("prompt_token_ids", T.TYPE_UINT32, REP, None),
        ("pooling", T.TYPE_STRING, OPT, None),
    ])
    msg("EmbedResponse", [
        ("values", T.TYPE_FLOAT, REP, None),
        ("prompt_tokens", T.TYPE_UINT32, OPT, None),
    ])
    msg("HealthRequest", [])
    msg("HealthResponse", [("ok", T.TYPE_BOOL, OPT, None)])

    pool = descriptor_pool.DescriptorPool()
    fd = pool.Add(fdp)
    return {
        name: message_factory.GetMessageClass(fd.message_types_by_name[name])
        for name in ("SamplingOptions", "GenerateRequest", "GenerateChunk",
                     "EmbedRequest", "EmbedResponse", "HealthRequest",
                     "HealthResponse")
    }


MSG = _build_messages()


def _sampling_params(req):
    from vllm_amd.sampling_params import RequestOutputKind, SamplingParams

    s = req.sampling
    # proto3 scalars have no presence: 0 means "unset" for the fields
    # whose zero value is not meaningful (temperature/top_p/max_tokens);
    # seed presence is explicit via has_seed.
    return SamplingParams(
        temperature=s.temperature or 1.0,
        top_p=s.top_p or 1.0,
        top_k=s.top_k,
        max_tokens=s.max_tokens or 16,
        seed=int(s.seed) if s.has_seed else None,
        ignore_eos=s.ignore_eos,
        stop=list(s.stop) or None,
        output_kind=RequestOutputKind.DELTA,
    )


class InferenceService:
    def __init__(self, engine, lora_names):
        self.engine = engine
        self.lora_names = lora_names
        self._counter = 0

    async def Generate(self, request, context):
        from vllm_amd.entrypoints.openai.protocol import random_id

        prompt = (list(request.prompt_token_ids)
                  if request.prompt_token_ids else request.prompt)
        params = _sampling_params(request)
        rid = request.request_id or random_id("grpc")
        lora = request.lora if request.lora in self.lora_names else None
        sent = 0
        async for out in self.engine.generate(prompt, params, rid,
                                              lora=lora):
            comp = out.outputs[0]
            # comp.token_ids is cumulative; the stream carries deltas
            # (comp.text is already a delta under RequestOutputKind.DELTA).
            delta = comp.token_ids[sent:]
            sent = len(comp.token_ids)
            yield MSG["GenerateChunk"](
                token_ids=delta,
                text=comp.text,
                finish_reason=comp.finish_reason or "",
                prompt_tokens=len(out.prompt_token_ids),
            )

    async def Embed(self, request, context):
        from vllm_amd.entrypoints.openai.protocol import random_id
        from vllm_amd.sampling_params import SamplingParams

        prompt = (list(request.prompt_token_ids)
                  if request.prompt_token_ids else request.prompt)
        params = SamplingParams(pooling=request.pooling or "last",
                                max_tokens=1)
        final = None
        async for out in self.engine.generate(prompt, params,
                                              random_id("grpc-embd")):
            final = out
        return MSG["EmbedResponse"](
            values=final.pooled or [],
            prompt_tokens=len(final.prompt_token_ids),
        )

    async def Health(self, request, context):
        return MSG["HealthResponse"](ok=True)


def make_grpc_server(engine, lora_names=(),
                     address: str = "127.0.0.1:0") -> tuple:
    """Build a grpc.aio server bound to `address`. Returns (server, port)
    — call await server.start() from a running loop."""
    service = InferenceService(engine, set(lora_names))
    handlers = {
        "Generate": grpc.unary_stream_rpc_method_handler(
            service.Generate,
            request_deserializer=MSG["GenerateRequest"].FromString,
            response_serializer=MSG["GenerateChunk"].SerializeToString,
        ),
        "Embed": grpc.unary_unary_rpc_method_handler(
            service.Embed,
            request_deserializer=MSG["EmbedRequest"].FromString,
            response_serializer=MSG["EmbedResponse"].SerializeToString,
        ),
        "Health": grpc.unary_unary_rpc_method_handler(
            service.Health,
            request_deserializer=MSG["HealthRequest"].FromString,
            response_serializer=MSG["HealthResponse"].SerializeToString,
        ),
    }
    server = grpc.aio.server()
    server.add_generic_rpc_handlers((
        grpc.method_handlers_generic_handler(f"{_PKG}.Inference", handlers),
    ))
    port = server.add_insecure_port(address)
    return server, port
