// xmlio.hpp — XML persistence for circuit states (gates.xsd format),
// checkpoint/resume compatible with reference-written files.
#pragma once

#include <string>

#include "sbg/state.hpp"

namespace sbg {

// Serializes st to the gates.xsd XML format (identical text format to the
// reference writer, state.c:107-166). Returns the XML text.
std::string state_to_xml(const state& st);

// Writes st to `dir`/<auto name> (see state_file_name). Returns the full
// path, or "" on failure. dir == "" means the current directory.
std::string save_state(const state& st, const std::string& dir = "");

// Parses and validates a gates.xsd XML document, recomputing every gate's
// truth table (truth tables are never serialized; parity: state.c:260-411,
// including the ordering/arity/range validation). Returns false on any
// validation error; *err receives a message when non-null.
bool state_from_xml(const std::string& xml, state* out, std::string* err);

// Loads a state from a file. Returns false on error.
bool load_state(const std::string& path, state* out, std::string* err);

}  // namespace sbg
