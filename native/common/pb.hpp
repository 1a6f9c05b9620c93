// pb.hpp — minimal protobuf wire-format writer.
//
// Companion to the reader in native/exporter/podresources.cpp: used by the
// OTLP exporter to emit binary-protobuf payloads
// (OTEL_EXPORTER_OTLP_PROTOCOL=http/protobuf — the OTLP/HTTP default
// encoding, required by collectors that do not enable the JSON codec).
// Nested messages are built bottom-up as byte strings and embedded
// length-delimited; no descriptors, no reflection.
#pragma once

#include <cstdint>
#include <string>

namespace pb {

inline void varint(std::string& out, uint64_t v) {
  while (v >= 0x80) {
    out += static_cast<char>((v & 0x7F) | 0x80);
    v >>= 7;
  }
  out += static_cast<char>(v);
}

inline void tag(std::string& out, uint32_t field, uint32_t wire) {
  varint(out, (static_cast<uint64_t>(field) << 3) | wire);
}

// field: varint scalar
inline void put_varint(std::string& out, uint32_t field, uint64_t v) {
  tag(out, field, 0);
  varint(out, v);
}

// field: 64-bit fixed (doubles, fixed64 timestamps)
inline void put_fixed64(std::string& out, uint32_t field, uint64_t v) {
  tag(out, field, 1);
  for (int i = 0; i < 8; i++) out += static_cast<char>((v >> (8 * i)) & 0xFF);
}

// field: length-delimited bytes/string/submessage
inline void put_bytes(std::string& out, uint32_t field, const std::string& data) {
  tag(out, field, 2);
  varint(out, data.size());
  out += data;
}

}  // namespace pb
