/* deepflow-amd shared-object plugin ABI (reference: agent
 * plugin/shared_obj — custom L7 protocol parsers loaded at runtime).
 *
 * A plugin implements:
 *   int df_plugin_parse(const uint8_t* payload, uint32_t len,
 *                       uint16_t server_port, DfPluginInfo* out);
 * returning 1 when it parsed the session (out filled), 0 to pass.
 * Compile: g++ -shared -fPIC myproto.cpp -o myproto.so
 */
#pragma once
#include <stdint.h>

typedef struct DfPluginInfo {
    char req_type[32];
    char domain[128];
    char resource[256];
    char endpoint[128];
    int32_t status;   /* 0 ok, 3 server error, 4 client error */
    int32_t code;
} DfPluginInfo;

#define DF_PLUGIN_PARSE_SYMBOL "df_plugin_parse"
