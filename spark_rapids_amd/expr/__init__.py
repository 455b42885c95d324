from .expressions import CaseWhen, Expression, col, lit, when
from .aggregates import AggExpr, avg, count, count_star, max_, min_, sum_
