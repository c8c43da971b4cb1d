SELECT abs(-3.5) AS a, sqrt(25.0) AS s;
SELECT floor(2.9) AS f, ceil(2.1) AS c, round(2.567, 1) AS r;
SELECT pow(3.0, 3.0) AS p;
SELECT 2 + 3 * 4 - 1 AS prec, (2 + 3) * 4 AS paren
