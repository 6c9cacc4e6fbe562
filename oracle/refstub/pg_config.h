/*
 * Minimal stand-in for configure-generated pg_config.h, ONLY for compiling
 * a few leaf files of /root/reference standalone into oracle/_ref/ (test
 * infrastructure — see oracle/Makefile).  Values mirror a standard x86-64
 * linux build of the reference (PostgreSQL 16.9 kernel, configure.ac:24).
 */
#define PG_VERSION_NUM 160009
#define ALIGNOF_DOUBLE 8
#define ALIGNOF_INT 4
#define ALIGNOF_LONG 8
#define ALIGNOF_SHORT 2
#define ALIGNOF_PG_INT128_TYPE 16
#define MAXIMUM_ALIGNOF 8
#define SIZEOF_VOID_P 8
#define SIZEOF_SIZE_T 8
#define SIZEOF_LONG 8
#define SIZEOF_OFF_T 8
#define SIZEOF_BOOL 1
#define HAVE_LONG_INT_64 1
#define PG_INT64_TYPE long int
#define PG_PRINTF_ATTRIBUTE printf
#define pg_restrict __restrict
#define PG_USE_STDBOOL 1
#define HAVE_STDBOOL_H 1
#define HAVE_STRINGS_H 1
#define HAVE_STDINT_H 1
#define HAVE_SYS_STAT_H 1
#define HAVE_UNISTD_H 1
#define HAVE_INTTYPES_H 1
#define HAVE_MEMORY_H 1
#define HAVE_STDLIB_H 1
#define HAVE_STRING_H 1
#define HAVE_SYS_TYPES_H 1
#define BLCKSZ 8192
#define XLOG_BLCKSZ 8192
#define NAMEDATALEN 64
#define FUNC_MAX_ARGS 100
#define INDEX_MAX_KEYS 32
#define PG_KRB_SRVNAM "postgres"
#define DEF_PGPORT 5432
#define DEF_PGPORT_STR "5432"
#define PACKAGE_NAME "PostgreSQL"
#define PACKAGE_VERSION "16.9"
#define PG_VERSION "16.9"
#define PG_MAJORVERSION "16"
#define PG_MAJORVERSION_NUM 16
#define PG_MINORVERSION_NUM 9
#define PG_VERSION_STR "PostgreSQL 16.9 oracle-ref stub"
#define MEMSET_LOOP_LIMIT 1024
#define USE_FLOAT8_BYVAL 1
#define INT64_MODIFIER "l"
