/* ===========================================================================
 * Restated declarations of the subset of Memgraph's module C ABI
 * (reference include/mg_procedure.h, 2194 lines) that the four drop-in
 * modules import. This header is written from scratch against the
 * reference's documented contract — each declaration cites the
 * mg_procedure.h line it mirrors. The HOST implements these symbols
 * (memgraphd exports only mgp_* to modules — include/mg_procedure.syms);
 * our modules only IMPORT them, and tests/mock/mgp_mock.cpp provides a
 * host-mock implementation for testing without a memgraphd.
 *
 * ABI facts mirrored exactly:
 *  - every fallible call returns `enum mgp_error` with out-parameters
 *    (mg_procedure.h:39-54: 14 enumerators, NO_ERROR == 0 first);
 *  - iterators/vertices created with an mgp_memory are procedure-scoped and
 *    must be destroyed by the module (mg_procedure.h:69-80 + destroy fns);
 *  - the read-procedure callback is
 *    void (*)(mgp_list*, mgp_graph*, mgp_result*, mgp_memory*)
 *    (mg_procedure.h:1843);
 *  - modules export int mgp_init_module(mgp_module*, mgp_memory*) and
 *    optionally int mgp_shutdown_module(void) (module.cpp:868,913).
 * ======================================================================== */
#ifndef MGX_MGP_H
#define MGX_MGP_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* mg_procedure.h:39-54 */
enum mgp_error {
  MGP_ERROR_NO_ERROR,
  MGP_ERROR_UNKNOWN_ERROR,
  MGP_ERROR_UNABLE_TO_ALLOCATE,
  MGP_ERROR_INSUFFICIENT_BUFFER,
  MGP_ERROR_OUT_OF_RANGE,
  MGP_ERROR_LOGIC_ERROR,
  MGP_ERROR_DELETED_OBJECT,
  MGP_ERROR_INVALID_ARGUMENT,
  MGP_ERROR_KEY_ALREADY_EXISTS,
  MGP_ERROR_IMMUTABLE_OBJECT,
  MGP_ERROR_VALUE_CONVERSION,
  MGP_ERROR_SERIALIZATION_ERROR,
  MGP_ERROR_AUTHORIZATION_ERROR,
  MGP_ERROR_NOT_YET_IMPLEMENTED,
};

/* Opaque host types (mg_procedure.h:146-180, 718-780, 1695+). */
struct mgp_memory;
struct mgp_value;
struct mgp_list;
struct mgp_vertex;
struct mgp_edge;
struct mgp_graph;
struct mgp_result;
struct mgp_result_record;
struct mgp_vertices_iterator;
struct mgp_edges_iterator;
struct mgp_module;
struct mgp_proc;
struct mgp_type;

/* mg_procedure.h:786-790 */
struct mgp_vertex_id {
  int64_t as_int;
};

/* --- values (mg_procedure.h:233-461) --- */
enum mgp_error mgp_value_make_bool(int val, struct mgp_memory *memory,
                                   struct mgp_value **result);              /* :233 */
enum mgp_error mgp_value_make_int(int64_t val, struct mgp_memory *memory,
                                  struct mgp_value **result);               /* :238 */
enum mgp_error mgp_value_make_double(double val, struct mgp_memory *memory,
                                     struct mgp_value **result);            /* :243 */
enum mgp_error mgp_value_make_string(const char *val, struct mgp_memory *memory,
                                     struct mgp_value **result);            /* :248 */
enum mgp_error mgp_value_make_vertex(struct mgp_vertex *val,
                                     struct mgp_value **result);            /* :272 */
void mgp_value_destroy(struct mgp_value *val);                              /* :~225 */
enum mgp_error mgp_value_is_null(struct mgp_value *val, int *result);       /* :357 */
enum mgp_error mgp_value_is_int(struct mgp_value *val, int *result);        /* :365 */
enum mgp_error mgp_value_is_double(struct mgp_value *val, int *result);     /* :369 */
enum mgp_error mgp_value_get_bool(struct mgp_value *val, int *result);      /* :431 */
enum mgp_error mgp_value_get_int(struct mgp_value *val, int64_t *result);   /* :436 */
enum mgp_error mgp_value_get_double(struct mgp_value *val, double *result); /* :441 */
enum mgp_error mgp_value_get_string(struct mgp_value *val, const char **result); /* :446 */
enum mgp_error mgp_value_get_list(struct mgp_value *val, struct mgp_list **result); /* :451 */
enum mgp_error mgp_value_get_vertex(struct mgp_value *val,
                                    struct mgp_vertex **result);            /* :461 */
enum mgp_error mgp_value_get_edge(struct mgp_value *val,
                                  struct mgp_edge **result);                /* :466 */

/* --- lists (mg_procedure.h:519-563) --- */
enum mgp_error mgp_list_make_empty(size_t capacity, struct mgp_memory *memory,
                                   struct mgp_list **result);               /* :519 */
enum mgp_error mgp_value_make_list(struct mgp_list *val,
                                   struct mgp_value **result);              /* :256 */
enum mgp_error mgp_list_append_extend(struct mgp_list *list,
                                      struct mgp_value *val);               /* :544 */
enum mgp_error mgp_list_size(struct mgp_list *list, size_t *result);        /* :554 */
enum mgp_error mgp_list_at(struct mgp_list *list, size_t index,
                           struct mgp_value **result);                      /* :563 */

/* enterprise gate used by the online modules (mg_procedure.h:65) */
int mgp_is_enterprise_valid(void);

/* --- results (mg_procedure.h:716-731) --- */
enum mgp_error mgp_result_set_error_msg(struct mgp_result *res,
                                        const char *error_msg);             /* :716 */
enum mgp_error mgp_result_new_record(struct mgp_result *res,
                                     struct mgp_result_record **result);    /* :721 */
enum mgp_error mgp_result_record_insert(struct mgp_result_record *record,
                                        const char *field_name,
                                        struct mgp_value *val);             /* :731 */

/* --- vertices & edges (mg_procedure.h:782-1007) --- */
void mgp_vertex_destroy(struct mgp_vertex *v);                              /* :845 */
enum mgp_error mgp_vertex_get_id(struct mgp_vertex *v,
                                 struct mgp_vertex_id *result);             /* :790 */
enum mgp_error mgp_vertex_iter_out_edges(struct mgp_vertex *v, struct mgp_memory *memory,
                                         struct mgp_edges_iterator **result); /* :901 */
void mgp_edges_iterator_destroy(struct mgp_edges_iterator *it);             /* :782 */
enum mgp_error mgp_edges_iterator_get(struct mgp_edges_iterator *it,
                                      struct mgp_edge **result);            /* :913 */
enum mgp_error mgp_edges_iterator_next(struct mgp_edges_iterator *it,
                                       struct mgp_edge **result);           /* :920 */
enum mgp_error mgp_edge_get_from(struct mgp_edge *e, struct mgp_vertex **result); /* :956 */
enum mgp_error mgp_edge_get_to(struct mgp_edge *e, struct mgp_vertex **result);   /* :961 */
enum mgp_error mgp_edge_get_property(struct mgp_edge *e, const char *property_name,
                                     struct mgp_memory *memory,
                                     struct mgp_value **result);            /* :967 */

/* --- graph (mg_procedure.h:1001-1205, 1689) --- */
enum mgp_error mgp_graph_get_vertex_by_id(struct mgp_graph *g, struct mgp_vertex_id id,
                                          struct mgp_memory *memory,
                                          struct mgp_vertex **result);      /* :1001 */
enum mgp_error mgp_graph_is_transactional(struct mgp_graph *graph, int *result); /* :1140 */
enum mgp_error mgp_graph_iter_vertices(struct mgp_graph *g, struct mgp_memory *memory,
                                       struct mgp_vertices_iterator **result); /* :1187 */
void mgp_vertices_iterator_destroy(struct mgp_vertices_iterator *it);       /* :1182 */
enum mgp_error mgp_vertices_iterator_get(struct mgp_vertices_iterator *it,
                                         struct mgp_vertex **result);       /* :1199 */
enum mgp_error mgp_vertices_iterator_next(struct mgp_vertices_iterator *it,
                                          struct mgp_vertex **result);      /* :1689 */
enum mgp_error mgp_graph_approximate_vertex_count(struct mgp_graph *graph,
                                                  size_t *result);          /* :1202 */
enum mgp_error mgp_graph_approximate_edge_count(struct mgp_graph *graph,
                                                size_t *result);            /* :1205 */

/* --- types (mg_procedure.h:1712-1765) --- */
enum mgp_error mgp_type_bool(struct mgp_type **result);                     /* :1712 */
enum mgp_error mgp_type_string(struct mgp_type **result);                   /* :1716 */
enum mgp_error mgp_type_int(struct mgp_type **result);                      /* :1720 */
enum mgp_error mgp_type_float(struct mgp_type **result);                    /* :1724 */
enum mgp_error mgp_type_node(struct mgp_type **result);                     /* :1749 */
enum mgp_error mgp_type_relationship(struct mgp_type **result);             /* :1756 */
enum mgp_error mgp_type_list(struct mgp_type *element_type,
                             struct mgp_type **result);                     /* :1765 */
enum mgp_error mgp_type_nullable(struct mgp_type *type,
                                 struct mgp_type **result);                 /* :1802 */

/* --- procedure registration (mg_procedure.h:1843-1952) --- */
typedef void (*mgp_proc_cb)(struct mgp_list *, struct mgp_graph *, struct mgp_result *,
                            struct mgp_memory *);                           /* :1843 */
enum mgp_error mgp_module_add_read_procedure(struct mgp_module *module, const char *name,
                                             mgp_proc_cb cb,
                                             struct mgp_proc **result);     /* :1862 */
enum mgp_error mgp_proc_add_arg(struct mgp_proc *proc, const char *name,
                                struct mgp_type *type);                     /* :1914 */
enum mgp_error mgp_proc_add_opt_arg(struct mgp_proc *proc, const char *name,
                                    struct mgp_type *type,
                                    struct mgp_value *default_value);       /* :1938 */
enum mgp_error mgp_proc_add_result(struct mgp_proc *proc, const char *name,
                                   struct mgp_type *type);                  /* :1952 */

/* --- misc (mg_procedure.h:136-142, 1986) --- */
enum mgp_error mgp_track_current_thread_allocations(struct mgp_graph *graph);   /* :136 */
enum mgp_error mgp_untrack_current_thread_allocations(struct mgp_graph *graph); /* :142 */
int mgp_must_abort(struct mgp_graph *graph);                                /* :1986 */

#ifdef __cplusplus
}
#endif

#endif /* MGX_MGP_H */
