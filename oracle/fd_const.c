/* fd_const.c — Pasta field constants (computed by tools/gen_fd_const.py from
 * the published moduli; see oracle/fd.h header for provenance and the
 * oracle-only usage rule). */
#include "fd.h"

const fd_ctx FD_P = {
    /* mod */ {0x992d30ed00000001ULL, 0x224698fc094cf91bULL, 0x0000000000000000ULL, 0x4000000000000000ULL},
    /* r2  */ {0x8c78ecb30000000fULL, 0xd7d30dbd8b0de0e7ULL, 0x7797a99bc3c95d18ULL, 0x096d41af7b9cb714ULL},
    /* r3  */ {0xf185a5993a9e10f9ULL, 0xf6a68f3b6ac5b1d1ULL, 0xdf8d1014353fd42cULL, 0x2ae309222d2d9910ULL},
    /* inv */ 0x992d30ecffffffffULL,
    /* root */ {0xbdad6fabd87ea32fULL, 0xea322bf2b7bb7584ULL, 0x362120830561f81aULL, 0x2bce74deac30ebdaULL},
    /* root_inv */ {0xf0b87c7db2ce91f6ULL, 0x84a0a1d8859f066fULL, 0xb4ed8e647196dad1ULL, 0x2cd5282c53116b5cULL},
    /* t_odd */ {0x094cf91b992d30edULL, 0x00000000224698fcULL, 0x0000000000000000ULL, 0x0000000040000000ULL},
    /* t1_2 */ {0x04a67c8dcc969877ULL, 0x0000000011234c7eULL, 0x0000000000000000ULL, 0x0000000020000000ULL},
};

const fd_ctx FD_Q = {
    /* mod */ {0x8c46eb2100000001ULL, 0x224698fc0994a8ddULL, 0x0000000000000000ULL, 0x4000000000000000ULL},
    /* r2  */ {0xfc9678ff0000000fULL, 0x67bb433d891a16e3ULL, 0x7fae231004ccf590ULL, 0x096d41af7ccfdaa9ULL},
    /* r3  */ {0x008b421c249dae4cULL, 0xe13bda50dba41326ULL, 0x88fececb8e15cb63ULL, 0x07dd97a06e6792c8ULL},
    /* inv */ 0x8c46eb20ffffffffULL,
    /* root */ {0xa70e2c1102b6d05fULL, 0x9bb97ea3c106f049ULL, 0x9e5c4dfd492ae26eULL, 0x2de6a9b8746d3f58ULL},
    /* root_inv */ {0x57eecda0a84b6836ULL, 0x4ad38b9084b8a80cULL, 0xf4c8f353124086c1ULL, 0x2235e1a7415bf936ULL},
    /* t_odd */ {0x0994a8dd8c46eb21ULL, 0x00000000224698fcULL, 0x0000000000000000ULL, 0x0000000040000000ULL},
    /* t1_2 */ {0x04ca546ec6237591ULL, 0x0000000011234c7eULL, 0x0000000000000000ULL, 0x0000000020000000ULL},
};
