"""sonata_grpc protobuf schema, built at import time.

No protoc / grpc_tools exist in this environment, so instead of generated
`*_pb2.py` the FileDescriptorProto is constructed programmatically with the
google.protobuf runtime and message classes come from message_factory.
The schema mirrors the reference interface definition
(crates/frontends/grpc/proto/sonata_grpc.proto) field-for-field, so the
wire format is compatible with existing sonata_grpc clients.
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

PACKAGE = "sonata_grpc"
SERVICE_NAME = "sonata_grpc.sonata_grpc"

_F = descriptor_pb2.FieldDescriptorProto

_SCALAR = {
    "string": _F.TYPE_STRING,
    "bytes": _F.TYPE_BYTES,
    "float": _F.TYPE_FLOAT,
    "uint32": _F.TYPE_UINT32,
    "int64": _F.TYPE_INT64,
    "bool": _F.TYPE_BOOL,
}


def _build_file() -> descriptor_pb2.FileDescriptorProto:
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "sonata_grpc.proto"
    f.package = PACKAGE
    f.syntax = "proto3"

    def enum(name, values):
        e = f.enum_type.add()
        e.name = name
        for i, v in enumerate(values):
            val = e.value.add()
            val.name = v
            val.number = i

    enum("SynthesisMode",
         ["MODE_UNSPECIFIED", "MODE_LAZY", "MODE_PARALLEL", "MODE_BATCHED"])
    enum("Quality",
         ["QUALITY_UNSPECIFIED", "QUALITY_X_LOW", "QUALITY_LOW",
          "QUALITY_MEDIUM", "QUALITY_HIGH"])

    def message(name, fields):
        m = f.message_type.add()
        m.name = name
        oneof_n = 0
        for num, fname, ftype, opts in fields:
            fd = m.field.add()
            fd.name = fname
            fd.number = num
            fd.label = _F.LABEL_OPTIONAL
            if opts == "map_int64_string":
                # map<int64,string> -> nested MapEntry message
                fd.label = _F.LABEL_REPEATED
                fd.type = _F.TYPE_MESSAGE
                fd.type_name = f".{PACKAGE}.{name}.{_camel(fname)}Entry"
                entry = m.nested_type.add()
                entry.name = f"{_camel(fname)}Entry"
                entry.options.map_entry = True
                k = entry.field.add()
                k.name, k.number, k.type = "key", 1, _F.TYPE_INT64
                k.label = _F.LABEL_OPTIONAL
                v = entry.field.add()
                v.name, v.number, v.type = "value", 2, _F.TYPE_STRING
                v.label = _F.LABEL_OPTIONAL
                continue
            if ftype in _SCALAR:
                fd.type = _SCALAR[ftype]
            elif ftype.startswith("enum:"):
                fd.type = _F.TYPE_ENUM
                fd.type_name = f".{PACKAGE}.{ftype[5:]}"
            else:
                fd.type = _F.TYPE_MESSAGE
                fd.type_name = f".{PACKAGE}.{ftype}"
            if opts == "optional":
                # proto3 explicit presence -> synthetic oneof
                oo = m.oneof_decl.add()
                oo.name = f"_{fname}"
                fd.oneof_index = oneof_n
                fd.proto3_optional = True
                oneof_n += 1
        return m

    message("Empty", [])
    message("Version", [(1, "version", "string", None)])
    message("VoiceIdentifier", [(1, "voice_id", "string", None)])
    message("SynthesisOptions", [
        (1, "speaker", "string", "optional"),
        (2, "length_scale", "float", "optional"),
        (3, "noise_scale", "float", "optional"),
        (4, "noise_w", "float", "optional"),
    ])
    message("AudioInfo", [
        (1, "sample_rate", "uint32", None),
        (2, "num_channels", "uint32", None),
        (3, "sample_width", "uint32", None),
    ])
    message("VoiceInfo", [
        (1, "voice_id", "string", None),
        (2, "synth_options", "SynthesisOptions", None),
        (3, "speakers", None, "map_int64_string"),
        (4, "audio", "AudioInfo", None),
        (5, "language", "string", "optional"),
        (6, "quality", "enum:Quality", "optional"),
        (7, "supports_streaming_output", "bool", "optional"),
    ])
    message("VoicePath", [(1, "config_path", "string", None)])
    message("SpeechArgs", [
        (1, "rate", "uint32", "optional"),
        (2, "volume", "uint32", "optional"),
        (3, "pitch", "uint32", "optional"),
        (4, "appended_silence_ms", "uint32", "optional"),
    ])
    message("Utterance", [
        (1, "voice_id", "string", None),
        (2, "text", "string", None),
        (3, "speech_args", "SpeechArgs", None),
        (4, "synthesis_mode", "enum:SynthesisMode", None),
    ])
    message("VoiceSynthesisOptions", [
        (1, "voice_id", "string", None),
        (2, "synthesis_options", "SynthesisOptions", None),
    ])
    message("SynthesisResult", [
        (1, "wav_samples", "bytes", None),
        (2, "rtf", "float", None),
    ])
    message("WaveSamples", [(1, "wav_samples", "bytes", None)])
    return f


def _camel(s: str) -> str:
    return "".join(p.capitalize() for p in s.split("_"))


_POOL = descriptor_pool.DescriptorPool()
_FILE = _POOL.Add(_build_file())

MESSAGES = {
    name: message_factory.GetMessageClass(_POOL.FindMessageTypeByName(
        f"{PACKAGE}.{name}"))
    for name in ["Empty", "Version", "VoiceIdentifier", "SynthesisOptions",
                 "AudioInfo", "VoiceInfo", "VoicePath", "SpeechArgs",
                 "Utterance", "VoiceSynthesisOptions", "SynthesisResult",
                 "WaveSamples"]
}

# RPC name -> (request type, response type, server_streaming)
RPCS = {
    "GetSonataVersion": ("Empty", "Version", False),
    "LoadVoice": ("VoicePath", "VoiceInfo", False),
    "GetVoiceInfo": ("VoiceIdentifier", "VoiceInfo", False),
    "GetSynthesisOptions": ("VoiceIdentifier", "SynthesisOptions", False),
    "SetSynthesisOptions": ("VoiceSynthesisOptions", "SynthesisOptions", False),
    "SynthesizeUtterance": ("Utterance", "SynthesisResult", True),
    "SynthesizeUtteranceRealtime": ("Utterance", "WaveSamples", True),
}

MODE_UNSPECIFIED, MODE_LAZY, MODE_PARALLEL, MODE_BATCHED = range(4)
QUALITY_VALUES = {"x_low": 1, "low": 2, "medium": 3, "high": 4}
