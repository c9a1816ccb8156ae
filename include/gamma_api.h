/*
 * gamma_api.h — C ABI of the Gamma engine, MI355X-native rebuild.
 *
 * This header re-declares, signature-for-signature, the 20 exported entry
 * points of the reference engine's C ABI
 * (reference: internal/engine/c_api/gamma_api.h:26-198) so that the Go
 * PartitionServer's cgo binding (reference:
 * internal/engine/sdk/go/gamma/gamma.go:11-14,160-188) can load this
 * library unchanged.
 *
 * Conventions (reference: internal/engine/c_api/gamma_api.cc and
 * c_api/api_data/response.cc:43-58):
 *  - engine handle: opaque void* returned by Init.
 *  - (const char*, int len) inputs are serialized buffers:
 *      Init/SetConfig        : JSON engine config  (gamma_api.cc:36-70)
 *      CreateTable           : FlatBuffers gamma_api.Table (idl/fbs/table.fbs)
 *      AddOrUpdateDoc        : FlatBuffers gamma_api.Doc   (idl/fbs/doc.fbs)
 *      Search/Query          : protobuf vearchpb.SearchRequest/QueryRequest
 *                              (internal/proto/router_grpc.proto:168-192)
 *  - (char **out, int *len) outputs are malloc'd by the engine and freed by
 *    the caller with free() (response.cc:51, gamma.go:180). Note the
 *    reference allocates CStatus.msg with new[] (gamma_api.cc:150) while the
 *    Go side frees it with C.free; this implementation uses malloc for every
 *    outward buffer, including CStatus.msg.
 *  - CStatus: code 0 = OK; msg is NULL on success, malloc'd otherwise and
 *    freed by the caller (gamma.go:186-188).
 *  - int returns: 0 ok; -2 = request killed / memory exceeded
 *    (gamma_index_ivfpq.cc:600-602, ps/engine/gammacb/reader.go:170).
 *  - Threading: Search may be called concurrently from arbitrary threads
 *    while one background thread calls AddOrUpdateDoc/BuildIndex
 *    (engine.cc:1108-1127); SetKillStatus arrives concurrently and must
 *    interrupt in-flight scans.
 */

#ifndef GAMMA_API_H_
#define GAMMA_API_H_

#ifdef __cplusplus
extern "C" {
#endif

struct CStatus {
  int code;
  char *msg;
};

/* reference: gamma_api.h:26 — create an engine from a JSON config string. */
void *Init(const char *config_str, int len);

/* reference: gamma_api.h:34 — destroy the engine. 0 ok, 1 failed. */
int Close(void *engine);

/* reference: gamma_api.h:43 — create a table from a FlatBuffers Table. */
struct CStatus CreateTable(void *engine, const char *table_str, int len);

/* reference: gamma_api.h:52 — add or update one FlatBuffers Doc. */
int AddOrUpdateDoc(void *engine, const char *doc_str, int len);

/* reference: gamma_api.h:61 — delete a doc by primary key. */
int DeleteDoc(void *engine, const char *docid, int docid_len);

/* reference: gamma_api.h:68 — engine status JSON (malloc'd out buffer). */
void GetEngineStatus(void *engine, char **status, int *len);

/* reference: gamma_api.h:70 — memory info JSON (malloc'd out buffer). */
void GetMemoryInfo(void *engine, char **memory_info, int *len);

/* reference: gamma_api.h:78 — get a doc by primary key (FlatBuffers Doc). */
int GetDocByID(void *engine, const char *docid, int docid_len, char **doc_str,
               int *len);

/* reference: gamma_api.h:89 — get a doc by internal docid. */
int GetDocByDocID(void *engine, int docid, char next, char **doc_str,
                  int *len);

/* reference: gamma_api.h:96 — train + build the vector index. */
int BuildIndex(void *engine);

/* reference: gamma_api.h:103 — rebuild the vector index. */
int RebuildIndex(void *engine, int drop_before_rebuild, int limit_cpu,
                 int describe);

/* reference: gamma_api.h:112 — dump engine state to the config path. */
int Dump(void *engine);

/* reference: gamma_api.h:120 — load engine state from the config path. */
int Load(void *engine);

/* reference: gamma_api.h:129 — vector search; request/response protobuf. */
struct CStatus Search(void *engine, const char *request_str, int req_len,
                      char **response_str, int *res_len);

/* reference: gamma_api.h:132 — scalar/doc query; protobuf QueryRequest. */
struct CStatus Query(void *engine, const char *request_str, int req_len,
                     char **response_str, int *res_len);

/* reference: gamma_api.h:141 — set cache sizes etc. from JSON. */
int SetConfig(void *engine, const char *config_str, int len);

/* reference: gamma_api.h:150 — get config JSON (malloc'd out buffer). */
int GetConfig(void *engine, char **config_str, int *len);

/* reference: gamma_api.h:152 — backup command dispatch. */
struct CStatus Backup(void *engine, int command);

/* reference: gamma_api.h:171 — add a named index over one or more fields. */
struct CStatus AddFieldIndexWithParams(
    void *engine, const char *index_name, int index_name_len,
    const char *const *field_names, const int *field_name_lens,
    int field_name_count, const char *index_type, int index_type_len,
    const char *index_params, int index_params_len);

/* reference: gamma_api.h:191 — remove a named index. */
struct CStatus RemoveFieldIndex(void *engine, const char *index_name,
                                int index_name_len);

/* reference: gamma_api.h:194 — set process-wide memory watermark (MB). */
void SetMemoryLimitConfig(int memory_limit);

/* reference: gamma_api.h:196 — mark a request killed; in-flight scans for
 * (request_id, partition_id) must stop and return -2. */
void SetKillStatus(const char *request_id, int partition_id, int reason);

/* reference: gamma_api.h:198 — clear a kill mark. */
void DeleteKillStatus(const char *request_id, int partition_id);

#ifdef __cplusplus
}
#endif

#endif /* GAMMA_API_H */
