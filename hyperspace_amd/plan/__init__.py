from .expr import (And, BinComp, Col, Expr, In, IsNotNull, Lit, Not, Or,
                   col, lit, parse_predicate)
from .nodes import (BucketUnionNode, Filter, IndexScan, Join, LogicalPlan,
                    Project, Scan)
