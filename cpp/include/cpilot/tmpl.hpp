// Config template engine: a focused reimplementation of the subset of Go
// text/template that ContainerPilot configs use, rendered against the
// process environment with missingkey=zero semantics.
//
// Supported: {{ .VAR }} lookups, {{- -}} whitespace trimming, variables
// ($i), pipelines (a | b | c), nested calls with parens, string/number
// literals, if/else/end, range/end (incl. `range $i := ...`), and the
// functions: default, env, split, join, replaceAll, regexReplaceAll, loop,
// printf.
// Parity: /root/reference/config/template/template.go:19-180.
#pragma once

#include <string>

namespace cpilot {

// Render `text` against the current process environment.
// Throws std::runtime_error on parse or execution errors.
std::string renderTemplate(const std::string& text);

}  // namespace cpilot
